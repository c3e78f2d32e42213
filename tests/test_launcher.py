"""Launcher (axonrun) behavior: rank env pinning, GPU pinning env,
fail-fast kill-tree on first failure, timeline env
plumbing.
"""
import json
import os
import subprocess
import sys
import tempfile

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(np_, script, extra_args=(), timeout=120, env=None):
    env = {**os.environ, **(env or {})}
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["AXON_SHUTDOWN_TIMEOUT"] = "3"  # keep fail-fast tests quick
    return subprocess.run(
        [sys.executable, "-m", "axon.runner.run", "-np", str(np_),
         *extra_args, sys.executable, "-c", script],
        capture_output=True, text=True, env=env, timeout=timeout)


def test_launcher_env_and_collective():
    script = (
        "import os, torch, axon as hvd\n"
        "hvd.init()\n"
        "assert hvd.size() == 2\n"
        "assert os.environ['HIP_VISIBLE_DEVICES'] == str(hvd.rank())\n"
        "out = hvd.allreduce(torch.ones(3) * (hvd.rank() + 1), average=True)\n"
        "assert abs(out[0].item() - 1.5) < 1e-6\n"
        "print('RANK_OK', hvd.rank())\n"
        "hvd.shutdown()\n")
    r = _run(2, script)
    assert r.returncode == 0, r.stderr
    assert "RANK_OK 0" in r.stdout and "RANK_OK 1" in r.stdout


def test_launcher_failfast():
    # rank 1 exits non-zero immediately; rank 0 would hang in a collective
    # forever — the launcher must kill it and propagate the failure code.
    script = (
        "import os, sys, torch, axon as hvd\n"
        "hvd.init()\n"
        "if hvd.rank() == 1:\n"
        "    sys.exit(7)\n"
        "try:\n"
        "    hvd.allreduce(torch.ones(3), average=True)\n"
        "except RuntimeError:\n"
        "    sys.exit(1)  # fail-fast surfaced as an exception: also fine\n"
        "import time; time.sleep(600)\n")
    r = _run(2, script, timeout=180)
    assert r.returncode != 0
    assert "terminating remaining ranks" in r.stderr


def test_launcher_no_pin():
    script = (
        "import os, axon as hvd\n"
        "hvd.init()\n"
        "assert 'HIP_VISIBLE_DEVICES' not in os.environ or "
        "os.environ.get('AXON_PINNED') is None\n"
        "hvd.shutdown()\n")
    r = _run(2, script, extra_args=("--no-pin",))
    assert r.returncode == 0, r.stderr


def test_timeline_written():
    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "tl.json")
        script = (
            "import torch, axon as hvd\n"
            "hvd.init()\n"
            "for i in range(3):\n"
            "    hvd.allreduce(torch.ones(8), average=True, name=f't{i}')\n"
            "hvd.shutdown()\n")
        r = _run(2, script, extra_args=("--timeline", path))
        assert r.returncode == 0, r.stderr
        f0 = os.path.join(d, "tl.0.json")
        assert os.path.exists(f0), os.listdir(d)
        content = open(f0).read().rstrip().rstrip(",")
        events = json.loads(content + "]")
        names = {e.get("name") for e in events}
        assert "NEGOTIATE" in names and "CPU_OP" in names
        # per-tensor lanes got labeled
        lanes = [e for e in events if e.get("name") == "thread_name"]
        assert any("allreduce.t0" in json.dumps(e) for e in lanes)


def test_timeline_runtime_start_stop():
    """hvd.start_timeline/stop_timeline: runtime-controlled window — only
    the op inside the window lands in the file."""
    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "rt.json")
        script = (
            "import torch, axon as hvd\n"
            "hvd.init()\n"
            "hvd.allreduce(torch.ones(4), name='before')\n"
            f"hvd.start_timeline({path!r})\n"
            "hvd.allreduce(torch.ones(4), name='inside')\n"
            "hvd.stop_timeline()\n"
            "hvd.allreduce(torch.ones(4), name='after')\n"
            "hvd.shutdown()\n")
        r = _run(2, script)
        assert r.returncode == 0, r.stderr
        f0 = os.path.join(d, "rt.0.json")
        assert os.path.exists(f0), os.listdir(d)
        content = open(f0).read().rstrip().rstrip(",")
        blob = json.dumps(json.loads(content + "]"))
        assert "allreduce.inside" in blob
        assert "allreduce.before" not in blob
        assert "allreduce.after" not in blob


def test_duplicate_inflight_name_rejected():
    """Two async ops with the same name in flight must error loudly, not
    corrupt the pending table. Race-free setup: rank 1 delays its first
    submission, so rank 0's op is GUARANTEED still pending (negotiation
    cannot finish without rank 1) when rank 0 re-submits the name."""
    script = (
        "import time, torch, axon as hvd\n"
        "hvd.init()\n"
        "if hvd.rank() == 0:\n"
        "    h1 = hvd.allreduce_async_(torch.ones(4), name='dup')\n"
        "    time.sleep(0.2)  # let the first submit drain into pending\n"
        "    try:\n"
        "        h2 = hvd.allreduce_async_(torch.ones(4), name='dup')\n"
        "        hvd.synchronize(h2)\n"
        "        raise SystemExit('second submit should have failed')\n"
        "    except RuntimeError as e:\n"
        "        assert 'unique name' in str(e) or 'in flight' in str(e), e\n"
        "else:\n"
        "    time.sleep(0.8)\n"
        "    h1 = hvd.allreduce_async_(torch.ones(4), name='dup')\n"
        "hvd.synchronize(h1)\n"
        "hvd.shutdown()\n")
    r = _run(2, script)
    assert r.returncode == 0, r.stderr


def test_shape_mismatch_reported():
    """Mismatched allreduce shapes across ranks -> clean error on every
    rank (controller validation), not a hang or corruption."""
    script = (
        "import os, torch, axon as hvd\n"
        "hvd.init()\n"
        "n = 4 if hvd.rank() == 0 else 5\n"
        "try:\n"
        "    hvd.allreduce(torch.ones(n), name='mm')\n"
        "    raise SystemExit('mismatch not detected')\n"
        "except RuntimeError as e:\n"
        "    assert 'mismatch' in str(e), e\n"
        "hvd.shutdown()\n")
    r = _run(2, script)
    assert r.returncode == 0, r.stderr


def test_stall_inspector_warns():
    """One rank submits, the other doesn't: the stall inspector must warn
    within AXON_STALL_CHECK_TIME."""
    script = (
        "import os, time, torch, axon as hvd\n"
        "hvd.init()\n"
        "if hvd.rank() == 0:\n"
        "    h = hvd.allreduce_async_(torch.ones(3), name='stalled')\n"
        "time.sleep(2.5)\n"
        "if hvd.rank() != 0:\n"
        "    h = hvd.allreduce_async_(torch.ones(3), name='stalled')\n"
        "hvd.synchronize(h)\n"
        "hvd.shutdown()\n")
    r = _run(2, script, env={"AXON_STALL_CHECK_TIME": "1"})
    assert r.returncode == 0, r.stderr
    assert "stall inspector" in r.stderr


def test_autotune_converges():
    """AXON_AUTOTUNE=1: the coordinator sweeps the (threshold, cycle) grid
    scored by allreduced bytes/s, ships knob updates in ResponseLists, and
    settles on a best sample (SURVEY.md §2.1 parameter manager)."""
    script = (
        "import torch, axon as hvd\n"
        "hvd.init()\n"
        "t = torch.ones(1 << 16)\n"
        "for i in range(1200):\n"
        "    hvd.allreduce(t, average=True, name=f'a{i % 7}')\n"
        "hvd.shutdown()\n")
    r = _run(2, script, env={"AXON_AUTOTUNE": "1", "AXON_AUTOTUNE_LOG": "1",
                             "AXON_AUTOTUNE_SAMPLE_SEC": "0.05",
                             "AXON_AUTOTUNE_WARMUP_SEC": "0.1",
                             "AXON_AUTOTUNE_REFINE": "0",
                             "AXON_AUTOTUNE_BO": "2",
                             "AXON_LOG_LEVEL": "info"}, timeout=150)
    assert r.returncode == 0, r.stderr
    assert "autotune sample" in r.stderr
    assert "autotune done" in r.stderr


def test_autotune_end_to_end_signal():
    """With a DistributedOptimizer stepping, the tuner scores steps/sec
    (end-to-end throughput) instead of allreduced bytes/sec, and runs a
    refinement round around the grid's best point."""
    script = (
        "import torch, axon as hvd\n"
        "hvd.init()\n"
        "m = torch.nn.Linear(64, 64)\n"
        "opt = hvd.DistributedOptimizer(torch.optim.SGD(m.parameters(),"
        " lr=0.01), named_parameters=m.named_parameters())\n"
        "hvd.broadcast_parameters(m.state_dict(), root_rank=0)\n"
        "x = torch.randn(8, 64)\n"
        "for i in range(900):\n"
        "    opt.zero_grad()\n"
        "    m(x).sum().backward()\n"
        "    opt.step()\n"
        "hvd.shutdown()\n")
    r = _run(2, script, env={"AXON_AUTOTUNE": "1", "AXON_AUTOTUNE_LOG": "1",
                             "AXON_AUTOTUNE_SAMPLE_SEC": "0.04",
                             "AXON_AUTOTUNE_WARMUP_SEC": "0.1",
                             "AXON_AUTOTUNE_REFINE": "1",
                             "AXON_LOG_LEVEL": "info"}, timeout=150)
    assert r.returncode == 0, r.stderr
    assert "steps/s" in r.stderr, r.stderr[-2000:]
    assert "autotune done" in r.stderr
    # refinement happened: more samples than the 12-point base grid
    assert r.stderr.count("autotune sample") > 12, \
        r.stderr.count("autotune sample")


def test_multi_host_rejected():
    """-H with a remote host fails fast with the single-node scope message
    (horovodrun's -H exists; silently running on one host would be worse
    than an error)."""
    r = subprocess.run(
        [sys.executable, "-m", "axon.runner.run", "-np", "2",
         "-H", "nodeA:4,nodeB:4", sys.executable, "-c", "pass"],
        capture_output=True, text=True, timeout=60)
    assert r.returncode != 0
    assert "multi-host launch is not supported" in r.stderr
    r2 = subprocess.run(
        [sys.executable, "-m", "axon.runner.run", "-np", "1", "--no-pin",
         "-H", "localhost:1", sys.executable, "-c",
         "import axon; axon.init(); axon.shutdown()"],
        capture_output=True, text=True, timeout=120,
        env={**os.environ, "PYTHONPATH": REPO})
    assert r2.returncode == 0, r2.stderr


def test_sweep_tool_cpu_smoke():
    """tools/sweep.py: one-point sweep end-to-end on CPU (the xGMI
    re-tuning deliverable's harness must stay runnable)."""
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "sweep.py"),
         "--model", "resnet50", "--gpus", "1", "--steps", "1",
         "--warmup", "0", "--batch", "1", "--thresholds", "32",
         "--cycles", "1"],
        capture_output=True, text=True, timeout=300,
        env={**os.environ, "PYTHONPATH": REPO})
    assert r.returncode == 0, r.stderr[-1500:]
    assert "best: thr=32MiB" in r.stdout, r.stdout


def test_output_filename_per_rank_logs():
    with tempfile.TemporaryDirectory() as d:
        base = os.path.join(d, "job")
        script = (
            "import axon as hvd\n"
            "hvd.init()\n"
            "print(f'hello from rank {hvd.rank()}', flush=True)\n"
            "hvd.shutdown()\n")
        r = subprocess.run(
            [sys.executable, "-m", "axon.runner.run", "-np", "2",
             "--no-pin", "--output-filename", base,
             sys.executable, "-c", script],
            capture_output=True, text=True, timeout=120,
            env={**os.environ, "PYTHONPATH": REPO})
        assert r.returncode == 0, r.stderr
        assert "hello from rank 0" in r.stdout  # rank 0 on the console
        with open(base + ".rank1.log") as f:
            assert "hello from rank 1" in f.read()
        assert "hello from rank 1" not in r.stdout


def test_timeline_report_tool():
    """tools/timeline_report.py parses a real ws=2 timeline and reports
    phase stats + comm-busy fraction."""
    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "tl.json")
        script = (
            "import torch, axon as hvd\n"
            "hvd.init()\n"
            "for i in range(4):\n"
            "    hvd.allreduce(torch.ones(64), name=f'r{i}')\n"
            "hvd.shutdown()\n")
        r = _run(2, script, extra_args=("--timeline", path))
        assert r.returncode == 0, r.stderr
        rep = subprocess.run(
            [sys.executable, os.path.join(REPO, "tools", "timeline_report.py"),
             "--json", os.path.join(d, "tl.0.json")],
            capture_output=True, text=True, timeout=60)
        assert rep.returncode == 0, rep.stderr
        doc = json.loads(rep.stdout)
        assert doc["phases"]["CPU_OP"]["count"] == 4
        assert doc["phases"]["NEGOTIATE"]["count"] == 4
        assert 0 <= doc["comm_busy_frac"] <= 1
        assert any(t["tensor"].endswith("r0") for t in doc["top_tensors"])


def test_autotune_bayesian_proposals():
    """The GP/EI proposal engine (phase 2 of the tuner): seeded with the
    coarse grid over a smooth score surface whose optimum lies OFF the
    grid, the loop must find a half-octave-accurate optimum within a few
    proposals (here: the very first proposal hits it)."""
    import math

    from axon import _core

    MiB = 1 << 20

    def score(thr, cyc):
        d = (math.log2(thr / MiB) - 7.5) ** 2 + (math.log2(cyc) - 0.5) ** 2
        return 100.0 - 10.0 * d  # optimum (181 MiB, 1.41 ms), off-grid

    obs = [(t * MiB, c, score(t * MiB, c))
           for t in (32, 64, 128, 256) for c in (0.5, 1.0, 2.0)]
    grid_best = max(o[2] for o in obs)
    for _ in range(8):
        p = _core._autotune_bo_propose(obs)
        if p is None:
            break
        obs.append((p[0], p[1], score(p[0], p[1])))
    best = max(obs, key=lambda o: o[2])
    assert best[2] > grid_best  # BO beat the grid
    assert abs(math.log2(best[0] / MiB) - 7.5) <= 0.5
    assert abs(math.log2(best[1]) - 0.5) <= 0.5
    # degenerate flat scores: no model, no proposal
    flat = [(64 * MiB, 1.0, 5.0), (128 * MiB, 1.0, 5.0)]
    assert _core._autotune_bo_propose(flat) is None


def test_autotune_bo_fuzz_invariants():
    """Fuzz the GP/EI proposal: for arbitrary observation sets the
    proposal is either None or an in-range lattice point that was not
    already observed; never NaN, never out of the knob envelope."""
    import math
    import random

    from axon import _core

    MiB = 1 << 20
    rng = random.Random(7)
    for trial in range(40):
        n = rng.randint(0, 12)
        obs = []
        for _ in range(n):
            thr = int(2 ** rng.uniform(3, 10) * MiB)
            cyc = 2 ** rng.uniform(-2, 2)
            score = rng.choice([rng.uniform(-1e9, 1e9),
                                rng.uniform(0, 1), 0.0])
            obs.append((thr, cyc, score))
        p = _core._autotune_bo_propose(obs)
        if p is None:
            continue
        thr, cyc = p
        assert 8 * MiB <= thr <= 1024 * MiB + MiB
        assert 0.24 <= cyc <= 4.01
        assert math.isfinite(cyc)
        for o in obs:  # never re-proposes an observed point
            assert not (abs(math.log2(max(1, o[0]) / thr)) < 1e-6 and
                        abs(math.log2(o[1] / cyc)) < 1e-6), (o, p)
