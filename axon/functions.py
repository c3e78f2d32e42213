"""State distribution helpers: broadcast_parameters,
broadcast_optimizer_state, broadcast_object, allgather_object.

Capability parity: horovod/torch/functions.py. These
define the checkpoint/resume contract: checkpoints are
plain rank-0 torch.save dicts; resume = rank-0 load + broadcast_parameters
+ broadcast_optimizer_state.
"""
import io

import cloudpickle
import torch

from axon import mpi_ops
from axon.basics import rank


def broadcast_parameters(params, root_rank=0, process_set=None):
    """Broadcast a state_dict or list of (name, tensor) pairs from
    root_rank to all ranks (rank-0 weights win at startup). With a
    process_set, broadcasts within the set only (call from members only)."""
    if isinstance(params, dict):
        items = sorted(params.items())
    else:
        items = list(params)
    handles = []
    for name, t in items:
        if t is None:
            continue
        if not torch.is_tensor(t):
            continue  # non-tensor state_dict entries (e.g. num_batches_tracked ints are tensors; skip others)
        handles.append(mpi_ops.broadcast_async_(t.data, root_rank,
                                                name=f"bparam.{name}",
                                                process_set=process_set))
    for h in handles:
        mpi_ops.synchronize(h)


def broadcast_object(obj, root_rank=0, name=None, process_set=None):
    """Broadcast an arbitrary picklable object (cloudpickle -> uint8 tensor
    -> broadcast)."""
    name = name or "broadcast_object"
    if rank() == root_rank:
        buf = cloudpickle.dumps(obj)
        sz = torch.tensor([len(buf)], dtype=torch.int64)
    else:
        sz = torch.zeros(1, dtype=torch.int64)
    mpi_ops.broadcast_(sz, root_rank, name=f"{name}.size",
                       process_set=process_set)
    n = int(sz.item())
    if rank() == root_rank:
        data = torch.frombuffer(bytearray(buf), dtype=torch.uint8).clone()
    else:
        data = torch.empty(n, dtype=torch.uint8)
    mpi_ops.broadcast_(data, root_rank, name=f"{name}.data",
                       process_set=process_set)
    if rank() == root_rank:
        return obj
    return cloudpickle.loads(data.numpy().tobytes())


def allgather_object(obj, name=None, process_set=None):
    """Gather an arbitrary picklable object from every rank (or set
    member); returns the list indexed by (set-relative) rank."""
    name = name or "allgather_object"
    buf = cloudpickle.dumps(obj)
    data = torch.frombuffer(bytearray(buf), dtype=torch.uint8).clone()
    gathered = mpi_ops.allgather(data, name=f"{name}.data",
                                 process_set=process_set)
    sizes = mpi_ops.allgather(torch.tensor([data.numel()]),
                              name=f"{name}.size", process_set=process_set)
    out, off = [], 0
    for s in sizes.tolist():
        out.append(cloudpickle.loads(gathered[off:off + s].numpy().tobytes()))
        off += s
    return out


def _walk_tensors(obj, path=()):
    """Deterministic traversal of tensors inside nested dict/list/tuple."""
    if torch.is_tensor(obj):
        yield path, obj
    elif isinstance(obj, dict):
        for k in sorted(obj.keys(), key=repr):
            yield from _walk_tensors(obj[k], path + (k,))
    elif isinstance(obj, (list, tuple)):
        for i, v in enumerate(obj):
            yield from _walk_tensors(v, path + (i,))


def _strip_tensors(obj):
    if torch.is_tensor(obj):
        return ("__axon_tensor__", str(obj.dtype).replace("torch.", ""),
                tuple(obj.shape), obj.device.type)
    if isinstance(obj, dict):
        return {k: _strip_tensors(v) for k, v in obj.items()}
    if isinstance(obj, list):
        return [_strip_tensors(v) for v in obj]
    if isinstance(obj, tuple):
        return tuple(_strip_tensors(v) for v in obj)
    return obj


def _rebuild_tensors(obj, device_hint=None):
    if isinstance(obj, tuple) and len(obj) == 4 and obj[0] == "__axon_tensor__":
        dtype = getattr(torch, obj[1])
        # CPU state tensors (e.g. Adam 'step') stay on CPU; device tensors
        # are rebuilt on this rank's corresponding parameter device.
        dev = obj[3] if obj[3] == "cpu" or device_hint is None else device_hint
        return torch.empty(obj[2], dtype=dtype, device=dev)
    if isinstance(obj, dict):
        return {k: _rebuild_tensors(v, device_hint) for k, v in obj.items()}
    if isinstance(obj, list):
        return [_rebuild_tensors(v, device_hint) for v in obj]
    if isinstance(obj, tuple):
        return tuple(_rebuild_tensors(v, device_hint) for v in obj)
    return obj


def broadcast_optimizer_state(optimizer, root_rank=0, process_set=None):
    """Broadcast optimizer state from root so all ranks start identically
    (SURVEY.md §3.3): the state-dict *structure* (param_groups, scalar
    entries) goes via broadcast_object; every tensor entry is then
    broadcast in-place in a deterministic order. Non-root ranks allocate
    missing state tensors on the device of the corresponding parameter."""
    from axon.optimizer import _DistributedOptimizer

    if isinstance(optimizer, _DistributedOptimizer) and optimizer._handles:
        optimizer.synchronize()

    sd = optimizer.state_dict()
    meta = _strip_tensors(sd)
    meta = broadcast_object(meta, root_rank, name="opt_state.meta",
                            process_set=process_set)

    if rank() != root_rank:
        # Map state ids -> parameter device so rebuilt tensors land on the
        # right GPU.
        id_to_dev = {}
        groups = optimizer.param_groups
        flat_params = [p for g in groups for p in g["params"]]
        for pid, p in enumerate(flat_params):
            id_to_dev[pid] = p.device
        rebuilt = {"param_groups": meta["param_groups"], "state": {}}
        for pid, pstate in meta.get("state", {}).items():
            dev = id_to_dev.get(pid if isinstance(pid, int) else int(pid))
            rebuilt["state"][pid] = _rebuild_tensors(pstate, dev)
        sd = rebuilt

    handles = []
    for path, t in _walk_tensors(sd.get("state", {})):
        nm = "opt_state." + ".".join(str(x) for x in path)
        handles.append(mpi_ops.broadcast_async_(t, root_rank, name=nm,
                                                process_set=process_set))
    for h in handles:
        mpi_ops.synchronize(h)

    if rank() != root_rank:
        optimizer.load_state_dict(sd)
