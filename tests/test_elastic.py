"""Elastic recovery: fault-injection through the real launcher (SURVEY.md
§5.3c): a worker is killed mid-training; the job must continue with the
surviving ranks from the last in-memory commit."""
import json
import os
import subprocess
import sys
import tempfile
import uuid

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SCRIPT = os.path.join(REPO, "tests", "elastic_script.py")


def _run_elastic(np_, extra, out):
    cmd = [sys.executable, "-m", "axon.runner.run", "-np", str(np_),
           "--elastic", "--no-pin", "-v", *extra,
           sys.executable, SCRIPT, out]
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["OMP_NUM_THREADS"] = "1"
    return subprocess.run(cmd, env=env, capture_output=True, text=True,
                          timeout=180)


def test_elastic_shrink_on_failure(tmp_path):
    """max-restarts 0: worker 2's death shrinks the world 3 -> 2; the
    survivors finish all steps with identical weights."""
    out = str(tmp_path / f"el_{uuid.uuid4().hex[:8]}")
    r = _run_elastic(3, ["--min-np", "2", "--max-restarts", "0"], out)
    assert r.returncode == 0, r.stderr[-2000:]
    assert os.path.exists(out + ".died")  # fault actually injected
    assert not os.path.exists(out + ".w2.json")  # evicted worker wrote nothing
    res = {}
    for w in (0, 1):
        with open(f"{out}.w{w}.json") as f:
            res[w] = json.load(f)
    assert res[0]["size"] == 2 and res[1]["size"] == 2
    assert res[0]["step"] == 12 and res[1]["step"] == 12
    assert abs(res[0]["wsum"] - res[1]["wsum"]) < 1e-9
    assert "generation 1" in r.stderr
    # Survivors fired their reset callbacks with the shrunken world size.
    for w in (0, 1):
        with open(f"{out}.reset.w{w}") as f:
            assert f.read().strip().splitlines() == ["2"]


def test_elastic_respawn_on_failure(tmp_path):
    """max-restarts 1: the dead worker is respawned, world stays 3, and
    the respawned (state-less) worker resyncs from rank 0."""
    out = str(tmp_path / f"el_{uuid.uuid4().hex[:8]}")
    r = _run_elastic(3, ["--min-np", "3", "--max-restarts", "1"], out)
    assert r.returncode == 0, r.stderr[-2000:]
    assert os.path.exists(out + ".died")
    res = {}
    for w in (0, 1, 2):
        with open(f"{out}.w{w}.json") as f:
            res[w] = json.load(f)
    for w in (0, 1, 2):
        assert res[w]["size"] == 3
        assert res[w]["step"] == 12
    assert abs(res[0]["wsum"] - res[2]["wsum"]) < 1e-9


def test_elastic_double_failure_shrinks(tmp_path):
    """Two workers die at the same step: one generation change handles the
    multi-death sweep, the world shrinks 4 -> 2, survivors finish."""
    out = str(tmp_path / f"el_{uuid.uuid4().hex[:8]}")
    cmd = [sys.executable, "-m", "axon.runner.run", "-np", "4",
           "--elastic", "--no-pin", "-v", "--min-np", "2",
           "--max-restarts", "0",
           sys.executable, os.path.join(REPO, "tests", "elastic_script2.py"),
           out]
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["OMP_NUM_THREADS"] = "1"
    r = subprocess.run(cmd, env=env, capture_output=True, text=True,
                       timeout=180)
    assert r.returncode == 0, r.stderr[-2000:]
    assert os.path.exists(out + ".died2") and os.path.exists(out + ".died3")
    res = {}
    for w in (0, 1):
        with open(f"{out}.w{w}.json") as f:
            res[w] = json.load(f)
    assert res[0]["size"] == 2 and res[1]["size"] == 2
    assert res[0]["step"] == 10 and res[1]["step"] == 10
    assert abs(res[0]["wsum"] - res[1]["wsum"]) < 1e-9


def test_elastic_clean_exit_excluded_from_generation(tmp_path):
    """A worker that exits 0 mid-job must be dropped from the next
    generation (ADVICE r1 regression): its peers fail on the broken
    channel, respawn, and finish as a 2-rank world — pre-fix they hung
    waiting for the finished worker to re-join."""
    out = str(tmp_path / f"el_{uuid.uuid4().hex[:8]}")
    script3 = os.path.join(REPO, "tests", "elastic_script3.py")
    cmd = [sys.executable, "-m", "axon.runner.run", "-np", "3",
           "--elastic", "--no-pin", "-v", "--min-np", "2",
           "--max-restarts", "1", sys.executable, script3, out]
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["OMP_NUM_THREADS"] = "1"
    env["AXON_SHUTDOWN_TIMEOUT"] = "3"
    r = subprocess.run(cmd, env=env, capture_output=True, text=True,
                       timeout=150)
    assert r.returncode == 0, r.stderr[-2000:]
    assert os.path.exists(out + ".done1")  # the clean exit happened
    for w in (0, 2):
        with open(f"{out}.w{w}.json") as f:
            res = json.load(f)
        assert res["size"] == 2, res
        assert res["step"] == 8, res
    assert not os.path.exists(out + ".w1.json")


SCRIPT4 = os.path.join(REPO, "tests", "elastic_script4.py")


def _run_discover(np_, out, flips, total=60):
    import time

    disc = out + ".target"
    with open(disc, "w") as f:
        f.write(str(np_))
    cmd = [sys.executable, "-m", "axon.runner.run", "-np", str(np_),
           "--elastic", "--no-pin", "-v", "--min-np", "1",
           "--max-restarts", "0", "--discover-cmd", f"cat {disc}",
           "--discover-interval", "0.3",
           sys.executable, SCRIPT4, out, str(total)]
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["OMP_NUM_THREADS"] = "1"
    p = subprocess.Popen(cmd, env=env, stdout=subprocess.PIPE,
                         stderr=subprocess.PIPE, text=True)
    for wait_marker, value in flips:
        # wait until training in the CURRENT world is demonstrably under
        # way before resizing (a flip during startup would just be picked
        # up as the boot generation, not a mid-training interrupt)
        deadline = time.time() + 90
        while not os.path.exists(wait_marker) and time.time() < deadline:
            time.sleep(0.1)
        assert os.path.exists(wait_marker), "training never started"
        with open(disc + ".tmp", "w") as f:
            f.write(str(value))
        os.replace(disc + ".tmp", disc)
    out_, err = p.communicate(timeout=180)
    return p.returncode, out_, err


def test_elastic_discovery_grow(tmp_path):
    """Discovery raises the target 2 -> 3 mid-run: a third worker is
    spawned, survivors re-join at a commit boundary
    (HostsUpdatedInterrupt), and every worker finishes in the ws=3
    world with identical weights."""
    out = str(tmp_path / f"gr_{uuid.uuid4().hex[:8]}")
    rc, _o, err = _run_discover(2, out, [(out + ".started.2", 3)])
    assert rc == 0, err[-2000:]
    res = {}
    for w in (0, 1, 2):
        with open(f"{out}.w{w}.json") as f:
            res[w] = json.load(f)
    for w in (0, 1, 2):
        assert res[w]["size"] == 3, res
        assert res[w]["step"] == 60
    assert res[0]["sizes"] == [2, 3]
    assert abs(res[0]["wsum"] - res[2]["wsum"]) < 1e-9
    assert "generation 1: size=3" in err


def test_elastic_discovery_shrink(tmp_path):
    """Discovery lowers the target 3 -> 2: the highest worker is evicted
    (not treated as a failure), survivors continue in the ws=2 world."""
    out = str(tmp_path / f"sh_{uuid.uuid4().hex[:8]}")
    rc, _o, err = _run_discover(3, out, [(out + ".started.3", 2)])
    assert rc == 0, err[-2000:]
    res = {}
    for w in (0, 1):
        with open(f"{out}.w{w}.json") as f:
            res[w] = json.load(f)
    assert not os.path.exists(f"{out}.w2.json")  # evicted mid-run
    for w in (0, 1):
        assert res[w]["size"] == 2
        assert res[w]["step"] == 60
    assert res[0]["sizes"] == [3, 2]
    assert abs(res[0]["wsum"] - res[1]["wsum"]) < 1e-9
