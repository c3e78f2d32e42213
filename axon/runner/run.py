"""axonrun — the single-node launcher.

Replaces the reference's horovodrun/mpirun + GCP provisioning stack
(SURVEY.md §2.2 "Runner / CLI", §3.4): forks N local ranks, pins one rank
per MI355X via HIP_VISIBLE_DEVICES, passes the unix-socket rendezvous path,
waits on children and kills the whole tree on the first failure
(fail-fast). No ssh, no cloud, no mpirun.

`--elastic` switches to generation-based membership (SURVEY.md §5.3c,
axon/elastic.py): rank failures respawn the worker (up to --max-restarts
per worker) or shrink the world (down to --min-np); survivors recover from
their last in-memory `state.commit()`.

Usage:
    axonrun -np 8 python train.py [args...]
    axonrun -np 8 --elastic --min-np 4 python train_elastic.py
    python -m axon.runner.run -np 8 python train.py
"""
import argparse
import json
import os
import signal
import subprocess
import sys
import tempfile
import time
import uuid


def _redirect_kwargs(args, rank):
    """--output-filename FILE -> ranks > 0 append stdout+stderr to
    FILE.rank<N>.log (horovodrun's per-rank log capture). Rank 0 stays on
    the console so single-line stdout contracts (e.g. the bench JSON
    line) are never swallowed."""
    if not args.output_filename or rank == 0:
        return {}
    f = open(f"{args.output_filename}.rank{rank}.log", "ab", buffering=0)
    return {"stdout": f, "stderr": subprocess.STDOUT}


def main(argv=None):
    parser = argparse.ArgumentParser(
        prog="axonrun", description="axon single-node launcher")
    parser.add_argument("-np", "--num-proc", type=int, required=True,
                        help="number of ranks (one per GPU)")
    parser.add_argument("-H", "--hosts", "--hostfile", default=None,
                        help="NOT SUPPORTED: this engine is single-node by "
                             "design (xGMI is the fabric); multi-host "
                             "launches are rejected with this message "
                             "instead of silently running on one host")
    parser.add_argument("--no-pin", action="store_true",
                        help="do not pin ranks to GPUs via HIP_VISIBLE_DEVICES")
    parser.add_argument("--timeline", default=None,
                        help="write a chrome-trace timeline to this path")
    parser.add_argument("--output-filename", default=None,
                        help="redirect each non-zero rank's stdout+stderr "
                             "to FILE.rank<N>.log (horovodrun parity); "
                             "rank 0 stays on the console")
    parser.add_argument("--fusion-threshold-mb", type=int, default=None,
                        help="override AXON_FUSION_THRESHOLD (MiB)")
    parser.add_argument("--cycle-time-ms", type=float, default=None,
                        help="override AXON_CYCLE_TIME (ms)")
    parser.add_argument("--verbose", "-v", action="store_true")
    parser.add_argument("--elastic", action="store_true",
                        help="generation-based membership: recover from rank "
                             "failures (see axon/elastic.py)")
    parser.add_argument("--min-np", type=int, default=1,
                        help="elastic: abort if the world would shrink below")
    parser.add_argument("--discover-cmd", default=None,
                        help="elastic world-size discovery: a shell command "
                             "printing the DESIRED number of workers; polled "
                             "every --discover-interval seconds, growth "
                             "spawns workers / shrink evicts the "
                             "highest-numbered ones at a new generation")
    parser.add_argument("--discover-interval", type=float, default=1.0)
    parser.add_argument("--max-restarts", type=int, default=1,
                        help="elastic: respawns allowed per worker before it "
                             "is evicted (world shrinks)")
    parser.add_argument("command", nargs=argparse.REMAINDER,
                        help="command to launch per rank")
    args = parser.parse_args(argv)

    if not args.command:
        parser.error("no command given")
    if args.hosts:
        only_local = all(h.split(":")[0] in ("localhost", "127.0.0.1")
                         for h in args.hosts.split(",") if h)
        if not only_local:
            parser.error(
                "-H/--hosts: multi-host launch is not supported — this "
                "engine targets ONE node of MI355X GPUs over xGMI "
                "(BASELINE single-node scope); run one job per node")
    if args.elastic:
        return run_elastic(args)
    np_ = args.num_proc
    sock = os.path.join(tempfile.gettempdir(), f"axon_{uuid.uuid4().hex[:12]}.sock")

    procs = []
    for r in range(np_):
        env = dict(os.environ)
        env["AXON_RANK"] = str(r)
        env["AXON_SIZE"] = str(np_)
        env["AXON_LOCAL_RANK"] = str(r)
        env["AXON_CONTROL_SOCKET"] = sock
        if not args.no_pin:
            env["HIP_VISIBLE_DEVICES"] = str(r)
            env["AXON_PINNED"] = "1"
        if args.timeline:
            env["AXON_TIMELINE"] = args.timeline
        if args.fusion_threshold_mb is not None:
            env["AXON_FUSION_THRESHOLD"] = str(args.fusion_threshold_mb << 20)
        if args.cycle_time_ms is not None:
            env["AXON_CYCLE_TIME"] = str(args.cycle_time_ms)
        if args.verbose:
            print(f"[axonrun] rank {r}: {' '.join(args.command)}", file=sys.stderr)
        procs.append(subprocess.Popen(args.command, env=env,
                                      start_new_session=True,
                                      **_redirect_kwargs(args, r)))

    def kill_all(sig=signal.SIGTERM):
        for p in procs:
            if p.poll() is None:
                try:
                    os.killpg(os.getpgid(p.pid), sig)
                except (ProcessLookupError, PermissionError):
                    pass

    def forward(signum, _frame):
        kill_all(signum)

    signal.signal(signal.SIGTERM, forward)
    signal.signal(signal.SIGINT, forward)

    exit_code = 0
    try:
        live = set(range(np_))
        while live:
            for r in sorted(live):
                rc = procs[r].poll()
                if rc is None:
                    continue
                live.discard(r)
                if rc != 0:
                    print(f"[axonrun] rank {r} exited with code {rc} — "
                          "terminating remaining ranks", file=sys.stderr)
                    exit_code = rc
                    kill_all()
                    deadline = time.time() + 10
                    for p in procs:
                        while p.poll() is None and time.time() < deadline:
                            time.sleep(0.1)
                    kill_all(signal.SIGKILL)
                    live = set()
                    break
            time.sleep(0.05)
    finally:
        kill_all(signal.SIGKILL)
        try:
            os.unlink(sock)
        except OSError:
            pass
    return exit_code


def run_elastic(args):
    np_ = args.num_proc
    base = os.path.join(tempfile.gettempdir(), f"axon_{uuid.uuid4().hex[:12]}")
    genfile = base + ".gen"
    gen = 0
    members = list(range(np_))  # worker ids, membership order = rank order
    restarts = {w: 0 for w in members}

    def write_genfile():
        doc = {"gen": gen, "size": len(members),
               "socket": f"{base}.g{gen}.sock",
               "ranks": {str(w): i for i, w in enumerate(members)}}
        tmp = genfile + ".tmp"
        with open(tmp, "w") as f:
            json.dump(doc, f)
        os.replace(tmp, genfile)
        if args.verbose:
            print(f"[axonrun] generation {gen}: size={len(members)} "
                  f"members={members}", file=sys.stderr)

    procs = {}

    def spawn(worker):
        env = dict(os.environ)
        env["AXON_WORKER_ID"] = str(worker)
        env["AXON_ELASTIC_GENFILE"] = genfile
        if not args.no_pin:
            env["HIP_VISIBLE_DEVICES"] = str(worker)
            env["AXON_PINNED"] = "1"
        if args.timeline:
            env["AXON_TIMELINE"] = args.timeline
        if args.fusion_threshold_mb is not None:
            env["AXON_FUSION_THRESHOLD"] = str(args.fusion_threshold_mb << 20)
        if args.cycle_time_ms is not None:
            env["AXON_CYCLE_TIME"] = str(args.cycle_time_ms)
        procs[worker] = subprocess.Popen(args.command, env=env,
                                         start_new_session=True,
                                         **_redirect_kwargs(args, worker))

    def kill_all(sig=signal.SIGTERM):
        for p in procs.values():
            if p.poll() is None:
                try:
                    os.killpg(os.getpgid(p.pid), sig)
                except (ProcessLookupError, PermissionError):
                    pass

    signal.signal(signal.SIGTERM, lambda s_, f: kill_all(s_))
    signal.signal(signal.SIGINT, lambda s_, f: kill_all(s_))

    write_genfile()
    for w in members:
        spawn(w)

    def probe_discovery():
        try:
            r = subprocess.run(args.discover_cmd, shell=True,
                               capture_output=True, text=True, timeout=5)
            return int(r.stdout.strip())
        except (ValueError, subprocess.SubprocessError) as e:
            if args.verbose:
                print(f"[axonrun] discovery probe failed: {e}",
                      file=sys.stderr)
            return None

    exit_code = 0
    evicted_by_discovery = []
    next_probe = time.time() + args.discover_interval
    next_worker_id = max(members) + 1
    try:
        while True:
            # Discovery-driven membership changes (hvd.elastic discovery
            # parity, single node): grow by spawning fresh worker ids,
            # shrink by evicting the highest-numbered members. Running
            # survivors pick the new generation up at their next
            # state.commit() (HostsUpdatedInterrupt).
            if args.discover_cmd and time.time() >= next_probe:
                next_probe = time.time() + args.discover_interval
                target = probe_discovery()
                if target is not None:
                    target = min(target, 32)  # engine rank masks are 32-bit
                if target is not None and target >= args.min_np and \
                        target != len(members):
                    if target > len(members):
                        new = list(range(next_worker_id,
                                         next_worker_id + target - len(members)))
                        next_worker_id += len(new)
                        members = members + new
                        for w in new:
                            restarts[w] = 0
                        gen += 1
                        write_genfile()
                        for w in new:
                            spawn(w)
                    else:
                        drop = members[target:]
                        members = members[:target]
                        gen += 1
                        write_genfile()  # published BEFORE the kill: the
                        # survivors' next commit sees it as a resize, not
                        # a failure
                        for w in drop:
                            evicted_by_discovery.append(w)
                            if procs[w].poll() is None:
                                try:
                                    os.killpg(os.getpgid(procs[w].pid),
                                              signal.SIGTERM)
                                except (ProcessLookupError, PermissionError):
                                    pass
            for w in list(evicted_by_discovery):
                if procs[w].poll() is not None:
                    evicted_by_discovery.remove(w)  # reaped

            running = [w for w in members if procs[w].poll() is None]
            done_ok = [w for w in members if procs[w].poll() == 0]
            dead = [w for w in members
                    if procs[w].poll() not in (None, 0)]
            if done_ok and not dead:
                if not running:
                    break  # every member finished cleanly
            if dead:
                for w in dead:
                    rc = procs[w].poll()
                    print(f"[axonrun] worker {w} exited with code {rc}",
                          file=sys.stderr)
                respawn, evict = [], []
                for w in dead:
                    if restarts[w] < args.max_restarts:
                        restarts[w] += 1
                        respawn.append(w)
                    else:
                        evict.append(w)
                # Drop evicted workers AND workers that already finished
                # cleanly: a new generation must not list a rank that will
                # never re-join (survivors would hang waiting for it).
                members = [w for w in members
                           if w not in evict and w not in done_ok]
                if len(members) < args.min_np:
                    print(f"[axonrun] world would shrink below --min-np "
                          f"{args.min_np} — aborting", file=sys.stderr)
                    exit_code = 1
                    kill_all()
                    break
                gen += 1
                write_genfile()
                for w in respawn:
                    spawn(w)
                if args.verbose and evict:
                    print(f"[axonrun] evicted workers {evict}", file=sys.stderr)
            time.sleep(0.1)
    finally:
        time.sleep(0.2)
        kill_all(signal.SIGKILL)
        for suffix in [".gen", ".gen.tmp"]:
            try:
                os.unlink(base + suffix)
            except OSError:
                pass
        for g in range(gen + 1):
            try:
                os.unlink(f"{base}.g{g}.sock")
            except OSError:
                pass
    return exit_code


if __name__ == "__main__":
    sys.exit(main())
