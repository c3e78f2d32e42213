"""Shared test infrastructure.

Multi-process CPU tests (the bulk of CI): fork `world_size`
processes, each initializes axon over a per-test unix control socket and
runs the test body; failures propagate with tracebacks. GPU tests are
marked @pytest.mark.gpu and only run on an MI355X box.
"""
import multiprocessing as mp
import os
import sys
import tempfile
import traceback
import uuid

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an AMD GPU (MI355X)")
    # Keep the pytest process itself single-threaded: run_distributed
    # forks children from it, and threads created by in-process tests
    # (torch intra-op pools, HTTP test clients) can leave locked mutexes
    # behind in the fork — the occasional 90s child deadlock. Children
    # re-assert this for themselves in _child.
    os.environ.setdefault("OMP_NUM_THREADS", "1")
    try:
        import torch

        torch.set_num_threads(1)
    except Exception:
        pass


def _child(rank, size, sock, fn, args, q):
    os.environ["AXON_RANK"] = str(rank)
    os.environ["AXON_SIZE"] = str(size)
    os.environ["AXON_LOCAL_RANK"] = str(rank)
    os.environ["AXON_CONTROL_SOCKET"] = sock
    os.environ.setdefault("AXON_START_TIMEOUT", "60")
    try:
        import torch

        # Forked children inherit libgomp state from the parent; any OMP
        # parallel region would then hang waiting on worker threads that do
        # not exist post-fork. Single-thread the child before its first
        # parallel region — these tests only touch tiny tensors.
        torch.set_num_threads(1)
        import axon as hvd

        hvd.init()
        fn(rank, size, *args)
        hvd.shutdown()
        q.put((rank, None))
        q.close()
        q.join_thread()  # flush the feeder thread before hard exit
        os._exit(0)
    except BaseException:
        try:
            q.put((rank, traceback.format_exc()))
            q.close()
            q.join_thread()
        except Exception:
            pass
        os._exit(1)


def run_distributed(fn, world_size=2, args=(), timeout=90):
    """Run fn(rank, size, *args) in `world_size` forked processes with axon
    initialized. Raises with the first failing rank's traceback."""
    ctx = mp.get_context("fork")
    sock = os.path.join(tempfile.gettempdir(),
                        f"axon_test_{uuid.uuid4().hex[:10]}.sock")
    q = ctx.Queue()
    procs = [ctx.Process(target=_child, args=(r, world_size, sock, fn, args, q))
             for r in range(world_size)]
    for p in procs:
        p.start()
    results = {}
    import queue as _qm
    deadline_err = None
    for _ in range(world_size):
        try:
            rank, err = q.get(timeout=timeout)
            results[rank] = err
        except _qm.Empty:
            deadline_err = f"timeout: only {len(results)}/{world_size} ranks reported"
            break
    for p in procs:
        p.join(timeout=10)
        if p.is_alive():
            p.terminate()
            p.join(timeout=5)
            if p.is_alive():
                p.kill()
    errs = [f"rank {r}:\n{e}" for r, e in sorted(results.items()) if e]
    if deadline_err and not errs:
        errs.append(deadline_err)
    if errs:
        raise AssertionError("distributed test failed:\n" + "\n".join(errs))


@pytest.fixture
def dist():
    return run_distributed
