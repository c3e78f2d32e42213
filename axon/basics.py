"""Process lifecycle + identity: init/shutdown/rank/size/local_rank.

Capability parity: horovod/torch/__init__.py + horovod/common/basics.py
(SURVEY.md §2.2). Rendezvous is single-node: a unix control socket whose
path comes from the launcher (AXON_CONTROL_SOCKET) or is derived from
torchrun's MASTER_PORT.
"""
import atexit
import os

import torch


def _env_int(*names, default=None):
    for n in names:
        v = os.environ.get(n)
        if v is not None and v != "":
            return int(v)
    return default


def _control_path():
    p = os.environ.get("AXON_CONTROL_SOCKET")
    if p:
        return p
    port = os.environ.get("MASTER_PORT")
    if port:
        return f"/tmp/axon_ctrl_{port}.sock"
    return f"/tmp/axon_ctrl_{os.getpid()}.sock"


def init():
    """Initialize the axon core. Reads rank/size from the launcher's env
    (AXON_* from axonrun, or RANK/WORLD_SIZE/LOCAL_RANK from torchrun-style
    launchers); defaults to a single-process world.

    Also selects this rank's GPU when one is visible: under axonrun's
    HIP_VISIBLE_DEVICES pinning each rank sees a single device 0; otherwise
    device = local_rank % device_count.
    """
    from axon import _core

    if _core.is_initialized():
        return
    rank = _env_int("AXON_RANK", "RANK", "OMPI_COMM_WORLD_RANK", default=0)
    size = _env_int("AXON_SIZE", "WORLD_SIZE", "OMPI_COMM_WORLD_SIZE", default=1)
    local_rank = _env_int("AXON_LOCAL_RANK", "LOCAL_RANK",
                          "OMPI_COMM_WORLD_LOCAL_RANK", default=rank)
    _core.init(rank, size, local_rank, _control_path())
    from axon.process_sets import _refresh_global
    _refresh_global()
    if torch.cuda.is_available():
        ndev = torch.cuda.device_count()
        torch.cuda.set_device(local_rank % ndev if ndev > 1 else 0)
    atexit.register(shutdown)


def shutdown():
    """Graceful shutdown: rendezvous with all ranks, stop the background
    loop. Safe to call more than once; registered atexit."""
    from axon import _core

    if _core.is_initialized():
        _core.shutdown()


def is_initialized():
    from axon import _core

    return _core.is_initialized()


def _ensure_init():
    from axon import _core

    if not _core.is_initialized():
        raise RuntimeError("axon has not been initialized — call axon.init() first")
    return _core


def rank():
    return _ensure_init().rank()


def size():
    return _ensure_init().size()


def local_rank():
    return _ensure_init().local_rank()


def local_size():
    # Single-node engine: every rank is local.
    return size()


def cross_rank():
    return 0


def cross_size():
    return 1


def start_timeline(file_path, mark_cycles=False):
    """Begin writing the chrome-trace timeline to `file_path` (rank suffix
    appended). hvd.start_timeline parity; AXON_TIMELINE env does the same
    from process start."""
    if mark_cycles:
        os.environ["AXON_TIMELINE_MARK_CYCLES"] = "1"
    _ensure_init().start_timeline(file_path)


def stop_timeline():
    """Close the active timeline file (hvd.stop_timeline parity)."""
    _ensure_init().stop_timeline()
