"""Process sets: collectives over subgroups of ranks, each with its own
RCCL communicator (capability parity: horovod/common/process_sets +
horovod/torch process_set= arguments, SURVEY.md §2.1 "Process sets").

Usage (must be called identically on ALL ranks, like every collective):

    odd = axon.add_process_set([1, 3])
    if odd.included():
        axon.allreduce(t, process_set=odd)

The registration is itself a collective: every rank allgathers the member
list, verifies agreement, and registers the set under the same id.
"""
import threading

from axon.basics import _ensure_init, rank, size

_lock = threading.Lock()


class ProcessSet:
    """A subgroup of ranks. `process_set_id` is None until registered via
    add_process_set (or for the pre-registered global_process_set, 0)."""

    def __init__(self, ranks):
        self.ranks = sorted({int(r) for r in ranks})
        self.process_set_id = None

    def size(self):
        return len(self.ranks)

    def rank(self):
        """This rank's index within the set (None if not a member)."""
        me = rank()
        return self.ranks.index(me) if me in self.ranks else None

    def included(self):
        return rank() in self.ranks

    def __repr__(self):
        return f"ProcessSet(id={self.process_set_id}, ranks={self.ranks})"


global_process_set = ProcessSet([])
global_process_set.process_set_id = 0


def _refresh_global():
    global_process_set.ranks = list(range(size()))


def resolve_id(process_set):
    """None -> 0 (global); ProcessSet -> its registered id; int -> itself."""
    if process_set is None:
        return 0
    if isinstance(process_set, int):
        return process_set
    if process_set.process_set_id is None:
        raise ValueError("process set not registered — call add_process_set "
                         "first (collectively, on every rank)")
    return process_set.process_set_id


def add_process_set(process_set):
    """Register a new process set. COLLECTIVE: every rank must call with the
    same member list, in the same order relative to other registrations.
    Accepts a ProcessSet or an iterable of ranks; returns the registered
    ProcessSet."""
    from axon.functions import allgather_object  # lazy: avoid import cycle

    ps = process_set if isinstance(process_set, ProcessSet) \
        else ProcessSet(process_set)
    if ps.process_set_id is not None:
        raise ValueError("process set already registered")
    if not ps.ranks:
        raise ValueError("process set needs at least one rank")
    all_lists = allgather_object(ps.ranks, name=f"axon.ps.add.{ps.ranks}")
    for r, lst in enumerate(all_lists):
        if lst != ps.ranks:
            raise ValueError(f"add_process_set mismatch: rank {r} passed "
                             f"{lst}, this rank passed {ps.ranks}")
    core = _ensure_init()
    with _lock:
        ps.process_set_id = core.ps_register(ps.ranks)
    return ps


def remove_process_set(process_set):
    """Invalidate a registered set (collective; its RCCL communicator is
    torn down at shutdown). Subsequent ops on it raise."""
    core = _ensure_init()
    pid = resolve_id(process_set)
    if pid == 0:
        raise ValueError("cannot remove the global process set")
    core.ps_remove(pid)
    if isinstance(process_set, ProcessSet):
        process_set.process_set_id = None


def process_set_ranks(process_set_id):
    core = _ensure_init()
    return list(core.ps_ranks(process_set_id))
