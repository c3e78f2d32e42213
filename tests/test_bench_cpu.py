"""bench.py contract smoke on CPU: ws=2 over the socket control plane.

Runs the driver's entry point end-to-end (init, broadcast_parameters,
hook-overlapped allreduce, barrier, max-over-ranks elapsed, one JSON
line on rank 0) with two CPU ranks, torchrun-style env. AXON_GRAPH=1 is
set on purpose: graph capture is GPU-only and must degrade to the eager
path silently when no GPU is present.
"""
import json
import os
import subprocess
import sys
import uuid

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_bench_ws2(extra_args, extra_env):
    port = str(20000 + (uuid.uuid4().int % 20000))
    procs = []
    for r in range(2):
        env = os.environ.copy()
        env.update({"RANK": str(r), "WORLD_SIZE": "2", "LOCAL_RANK": "0",
                    "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": port})
        env.update(extra_env)
        procs.append(subprocess.Popen(
            [sys.executable, "bench.py", "--steps", "1", "--warmup", "1",
             "--batch", "2", "--image-size", "64"] + extra_args,
            cwd=REPO, env=env, stdout=subprocess.PIPE,
            stderr=subprocess.STDOUT, text=True))
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=600)
        outs.append(out)
        assert p.returncode == 0, f"bench rank failed:\n{out}"
    payloads = [line for out in outs for line in out.splitlines()
                if line.startswith("{")]
    assert len(payloads) == 1, f"expected one rank-0 JSON line:\n{outs}"
    return json.loads(payloads[0])


def test_bench_resnet_cpu_ws2():
    rec = _run_bench_ws2([], {"AXON_GRAPH": "1"})
    assert rec["n_gpus"] == 2
    assert rec["value"] > 0
    assert rec["config"]["parallelism"] == "dp2"
    assert rec["config"]["global_batch"] == 4
    assert rec["config"]["reduction"] == "average"


def test_bench_resnet_adasum_cpu_ws2():
    """Config-5 launch shape: the driver's bench contract with
    --reduction adasum runs end-to-end (engine Adasum path at ws=2)."""
    rec = _run_bench_ws2(["--reduction", "adasum"], {})
    assert rec["config"]["reduction"] == "adasum"
    assert rec["value"] > 0


def test_bench_resnet_cpu_ws8():
    """The scaling run's max launch shape: 8 ranks through the full bench
    stack (socket star with 8 connections, broadcast_parameters, hook
    allreduces, barrier, max-over-ranks elapsed) on CPU."""
    port = str(20000 + (uuid.uuid4().int % 20000))
    procs = []
    for r in range(8):
        env = os.environ.copy()
        env.update({"RANK": str(r), "WORLD_SIZE": "8", "LOCAL_RANK": "0",
                    "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": port,
                    "OMP_NUM_THREADS": "1"})
        procs.append(subprocess.Popen(
            [sys.executable, "bench.py", "--steps", "1", "--warmup", "0",
             "--batch", "1", "--image-size", "64"],
            cwd=REPO, env=env, stdout=subprocess.PIPE,
            stderr=subprocess.STDOUT, text=True))
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=600)
        outs.append(out)
        assert p.returncode == 0, f"bench rank failed:\n{out}"
    payloads = [line for out in outs for line in out.splitlines()
                if line.startswith("{")]
    assert len(payloads) == 1
    rec = json.loads(payloads[0])
    assert rec["n_gpus"] == 8
    assert rec["config"]["global_batch"] == 8


def test_bench_json_contract():
    """The driver parses one JSON line with these exact keys; pin them so
    a future edit can't silently break the contract."""
    port = str(20000 + (uuid.uuid4().int % 20000))
    env = os.environ.copy()
    env.update({"RANK": "0", "WORLD_SIZE": "1", "LOCAL_RANK": "0",
                "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": port})
    r = subprocess.run(
        [sys.executable, "bench.py", "--steps", "1", "--warmup", "0",
         "--batch", "1", "--image-size", "64"],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stdout + r.stderr
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][0]
    rec = json.loads(line)
    required = {"metric": str, "value": (int, float), "unit": str,
                "n_gpus": int, "steps": int, "warmup": int,
                "ms_per_step": (int, float), "higher_is_better": bool,
                "scaling": str, "dtype": str, "data": str, "config": dict}
    for k, t in required.items():
        assert k in rec and isinstance(rec[k], t), (k, rec.get(k))
    assert "vs_baseline" in rec  # null allowed (no published baseline)
    assert rec["scaling"] == "weak"
    assert rec["data"] == "synthetic"
    assert rec["config"]["model"] == "resnet50"
    assert "global_batch" in rec["config"]


def test_bench_under_real_torchrun_ws2():
    """The driver's exact launch form: torch.distributed.run --nnodes=1
    --nproc-per-node N --master-addr 127.0.0.1 ... bench.py."""
    port = str(20000 + (uuid.uuid4().int % 20000))
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", port, "bench.py", "--gpus", "2", "--steps", "1",
         "--warmup", "0", "--batch", "1", "--image-size", "64"],
        cwd=REPO, env=os.environ.copy(), capture_output=True, text=True,
        timeout=600)
    assert r.returncode == 0, (r.stdout + r.stderr)[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][0]
    rec = json.loads(line)
    assert rec["n_gpus"] == 2 and rec["config"]["parallelism"] == "dp2"
