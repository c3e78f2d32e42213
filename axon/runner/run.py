"""axonrun — the single-node launcher.

Replaces the reference's horovodrun/mpirun + GCP provisioning stack
(SURVEY.md §2.2 "Runner / CLI", §3.4): forks N local ranks, pins one rank
per MI355X via HIP_VISIBLE_DEVICES, passes the unix-socket rendezvous path,
waits on children and kills the whole tree on the first failure
(fail-fast). No ssh, no cloud, no mpirun.

Usage:
    axonrun -np 8 python train.py [args...]
    python -m axon.runner.run -np 8 python train.py
"""
import argparse
import os
import signal
import subprocess
import sys
import tempfile
import time
import uuid


def main(argv=None):
    parser = argparse.ArgumentParser(
        prog="axonrun", description="axon single-node launcher")
    parser.add_argument("-np", "--num-proc", type=int, required=True,
                        help="number of ranks (one per GPU)")
    parser.add_argument("--no-pin", action="store_true",
                        help="do not pin ranks to GPUs via HIP_VISIBLE_DEVICES")
    parser.add_argument("--timeline", default=None,
                        help="write a chrome-trace timeline to this path")
    parser.add_argument("--fusion-threshold-mb", type=int, default=None,
                        help="override AXON_FUSION_THRESHOLD (MiB)")
    parser.add_argument("--cycle-time-ms", type=float, default=None,
                        help="override AXON_CYCLE_TIME (ms)")
    parser.add_argument("--verbose", "-v", action="store_true")
    parser.add_argument("command", nargs=argparse.REMAINDER,
                        help="command to launch per rank")
    args = parser.parse_args(argv)

    if not args.command:
        parser.error("no command given")
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
                                      start_new_session=True))

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


if __name__ == "__main__":
    sys.exit(main())
