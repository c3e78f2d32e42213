"""Tensor collective API: the 4-way (sync/async x in-place/out-of-place)
op families plus synchronize/poll/join/barrier.

Capability parity: horovod/torch/mpi_ops.py (SURVEY.md §2.2 "Tensor ops
API"). Ops are enqueued to the C++ core (axon/_core.so) which coordinates
order across ranks and runs RCCL on a dedicated HIP stream.
"""
import threading
import time

import torch

from axon import metrics
from axon.basics import _ensure_init, rank, size
from axon.process_sets import resolve_id


class ReduceOp:
    Average = 0
    Sum = 1
    Adasum = 2
    Min = 3
    Max = 4
    Product = 5


# Horovod-compatible module-level aliases.
Average = ReduceOp.Average
Sum = ReduceOp.Sum
Adasum = ReduceOp.Adasum
Min = ReduceOp.Min
Max = ReduceOp.Max
Product = ReduceOp.Product

_lock = threading.Lock()
_counters = {}
_outputs = {}  # handle -> result payload returned by synchronize


def _auto_name(kind, name, psid=0):
    if name is not None:
        base = f"{kind}.{name}"
    else:
        # Deterministic per-op counter: every rank enqueues ops in the same
        # program order, so the counter-derived names agree across ranks.
        with _lock:
            n = _counters.get(kind, 0)
            _counters[kind] = n + 1
        base = f"{kind}.noname.{n}"
    # Scope names per process set so the same user name can be in flight on
    # different sets without colliding in the controller table.
    return base if psid == 0 else f"ps{psid}:{base}"


def _register(handle, payload, kind=None):
    with _lock:
        _outputs[handle] = (payload, kind, time.perf_counter())
    return handle


def _wire(compression):
    wd = getattr(compression, "wire_dtype", None) if compression is not None else None
    if wd is None:
        return -1
    return {torch.float16: 5, torch.bfloat16: 15, torch.float32: 6}[wd]


def _dense(t):
    """Dense (non-overlapping) tensors of ANY memory format (e.g.
    channels_last) participate in elementwise collectives directly: every
    rank constructs identical strides, so storage order agrees."""
    if t.is_contiguous():
        return True
    try:
        return torch.ops.aten.is_non_overlapping_and_dense(t)
    except Exception:
        return t.is_contiguous(memory_format=torch.channels_last) or \
            t.is_contiguous(memory_format=torch.channels_last_3d)


def _resolve_op(op, average):
    if op is not None and average is not None:
        raise ValueError("specify either op or average, not both")
    if op is None:
        if average is None or average:
            return ReduceOp.Average
        return ReduceOp.Sum
    return op


# --------------------------------------------------------------- allreduce

def allreduce_async(tensor, average=None, name=None, op=None,
                    prescale_factor=1.0, postscale_factor=1.0,
                    compression=None, process_set=None):
    """Out-of-place async allreduce; returns a handle."""
    red = _resolve_op(op, average)
    psid = resolve_id(process_set)
    t = tensor if _dense(tensor) else tensor.contiguous()
    output = torch.empty_like(t)
    metrics.record_op('allreduce', t.numel() * t.element_size())
    core = _ensure_init()
    h = core.allreduce_async(t, output, _auto_name("allreduce", name, psid),
                             red, prescale_factor, postscale_factor,
                             _wire(compression), psid)
    return _register(h, output, 'allreduce')


class _DiffAllreduce(torch.autograd.Function):
    """Differentiable allreduce (hvd parity): backward re-enters the same
    collective with the same op, so gradients of cross-rank expressions
    aggregate correctly (sum over every rank's loss)."""

    @staticmethod
    def forward(ctx, tensor, red, name, prescale, postscale, compression,
                process_set):
        ctx.args = (red, name, prescale, postscale, compression, process_set)
        return synchronize(allreduce_async(tensor, None, name, red, prescale,
                                           postscale, compression,
                                           process_set))

    @staticmethod
    def backward(ctx, grad):
        red, name, prescale, postscale, compression, process_set = ctx.args
        g = synchronize(allreduce_async(
            grad.contiguous(), None, f"{name}.grad" if name else None, red,
            prescale, postscale, compression, process_set))
        return (g,) + (None,) * 6


def allreduce(tensor, average=None, name=None, op=None, prescale_factor=1.0,
              postscale_factor=1.0, compression=None, process_set=None):
    if torch.is_grad_enabled() and tensor.requires_grad:
        red = _resolve_op(op, average)
        return _DiffAllreduce.apply(tensor, red, name, prescale_factor,
                                    postscale_factor, compression,
                                    process_set)
    return synchronize(allreduce_async(tensor, average, name, op,
                                       prescale_factor, postscale_factor,
                                       compression, process_set))


def allreduce_async_(tensor, average=None, name=None, op=None,
                     prescale_factor=1.0, postscale_factor=1.0,
                     compression=None, process_set=None):
    """In-place async allreduce (the hot path for gradients)."""
    if not _dense(tensor):
        raise ValueError("allreduce_ requires a dense tensor")
    red = _resolve_op(op, average)
    psid = resolve_id(process_set)
    metrics.record_op('allreduce', tensor.numel() * tensor.element_size())
    core = _ensure_init()
    h = core.allreduce_async(tensor, tensor,
                             _auto_name("allreduce", name, psid), red,
                             prescale_factor, postscale_factor,
                             _wire(compression), psid)
    return _register(h, tensor, 'allreduce')


def allreduce_(tensor, average=None, name=None, op=None, prescale_factor=1.0,
               postscale_factor=1.0, compression=None, process_set=None):
    return synchronize(allreduce_async_(tensor, average, name, op,
                                        prescale_factor, postscale_factor,
                                        compression, process_set))


def grouped_allreduce_async(tensors, average=None, name=None, op=None,
                            prescale_factor=1.0, postscale_factor=1.0,
                            compression=None, process_set=None):
    """Allreduce a list of tensors as one logical group. The group is
    negotiated in a single cycle, so the controller fuses it into as few
    collectives as the fusion threshold allows."""
    gname = _auto_name("grouped", name)
    return [allreduce_async(t, average, f"{gname}.{i}", op, prescale_factor,
                            postscale_factor, compression, process_set)
            for i, t in enumerate(tensors)]


def grouped_allreduce(tensors, **kw):
    if torch.is_grad_enabled() and any(t.requires_grad for t in tensors):
        # Differentiable path: one allreduce() per tensor; same-cycle
        # submissions still fuse in the controller like the async group.
        gname = _auto_name("grouped", kw.pop("name", None))
        return [allreduce(t, name=f"{gname}.{i}", **kw)
                for i, t in enumerate(tensors)]
    return [synchronize(h) for h in grouped_allreduce_async(tensors, **kw)]


def grouped_allreduce_async_(tensors, average=None, name=None, op=None,
                             prescale_factor=1.0, postscale_factor=1.0,
                             compression=None, process_set=None):
    gname = _auto_name("grouped", name)
    return [allreduce_async_(t, average, f"{gname}.{i}", op, prescale_factor,
                             postscale_factor, compression, process_set)
            for i, t in enumerate(tensors)]


def grouped_allreduce_(tensors, **kw):
    return [synchronize(h) for h in grouped_allreduce_async_(tensors, **kw)]


# --------------------------------------------------------------- broadcast

def broadcast_async(tensor, root_rank, name=None, process_set=None):
    psid = resolve_id(process_set)
    t = tensor if _dense(tensor) else tensor.contiguous()
    output = torch.empty_like(t)
    metrics.record_op('broadcast', t.numel() * t.element_size())
    core = _ensure_init()
    h = core.broadcast_async(t, output, root_rank,
                             _auto_name("broadcast", name, psid), psid)
    return _register(h, output, 'broadcast')


class _DiffBroadcast(torch.autograd.Function):
    @staticmethod
    def forward(ctx, tensor, root_rank, name, process_set):
        ctx.args = (root_rank, name, process_set)
        return synchronize(broadcast_async(tensor, root_rank, name,
                                           process_set))

    @staticmethod
    def backward(ctx, grad):
        root_rank, name, process_set = ctx.args
        # Output is root's tensor everywhere, so d(input) sums every rank's
        # grad on the root and is zero elsewhere.
        g = synchronize(allreduce_async(
            grad.contiguous(), None, f"{name}.grad" if name else None,
            ReduceOp.Sum, 1.0, 1.0, None, process_set))
        if rank() != root_rank:
            g = torch.zeros_like(g)
        return g, None, None, None


def broadcast(tensor, root_rank, name=None, process_set=None):
    if torch.is_grad_enabled() and tensor.requires_grad:
        return _DiffBroadcast.apply(tensor, root_rank, name, process_set)
    return synchronize(broadcast_async(tensor, root_rank, name, process_set))


def broadcast_async_(tensor, root_rank, name=None, process_set=None):
    if not _dense(tensor):
        raise ValueError("broadcast_ requires a dense tensor")
    psid = resolve_id(process_set)
    metrics.record_op('broadcast', tensor.numel() * tensor.element_size())
    core = _ensure_init()
    h = core.broadcast_async(tensor, tensor, root_rank,
                             _auto_name("broadcast", name, psid), psid)
    return _register(h, tensor, 'broadcast')


def broadcast_(tensor, root_rank, name=None, process_set=None):
    return synchronize(broadcast_async_(tensor, root_rank, name, process_set))


# --------------------------------------------------------------- allgather

def allgather_async(tensor, name=None, process_set=None):
    psid = resolve_id(process_set)
    t = tensor.contiguous()
    output = t.new_empty(0)
    metrics.record_op('allgather', t.numel() * t.element_size())
    core = _ensure_init()
    h = core.allgather_async(t, output, _auto_name("allgather", name, psid),
                             psid)
    return _register(h, output, 'allgather')


class _DiffAllgather(torch.autograd.Function):
    @staticmethod
    def forward(ctx, tensor, name, process_set):
        ctx.args = (name, process_set)
        ctx.dim0 = tensor.shape[0] if tensor.dim() > 0 else tensor.numel()
        return synchronize(allgather_async(tensor, name, process_set))

    @staticmethod
    def backward(ctx, grad):
        name, process_set = ctx.args
        gname = f"{name}.grad" if name else None
        # Sum every rank's grad of the (identical) gathered output, then
        # take this rank's slice (hvd semantics).
        g = synchronize(allreduce_async(grad.contiguous(), None, gname,
                                        ReduceOp.Sum, 1.0, 1.0, None,
                                        process_set))
        dims = synchronize(allgather_async(
            torch.tensor([ctx.dim0]), f"{gname}.dims" if gname else None,
            process_set))
        psid = resolve_id(process_set)
        my = rank() if psid == 0 else \
            list(_ensure_init().ps_ranks(psid)).index(rank())
        offset = int(dims[:my].sum())
        return g.narrow(0, offset, ctx.dim0), None, None


def allgather(tensor, name=None, process_set=None):
    if torch.is_grad_enabled() and tensor.requires_grad:
        return _DiffAllgather.apply(tensor, name, process_set)
    return synchronize(allgather_async(tensor, name, process_set))


# ---------------------------------------------------------------- alltoall

def alltoall_async(tensor, splits=None, name=None, process_set=None):
    psid = resolve_id(process_set)
    t = tensor.contiguous()
    n = size() if psid == 0 else len(_ensure_init().ps_ranks(psid))
    if splits is None:
        rows = t.shape[0] if t.dim() > 0 else t.numel()
        if rows % n != 0:
            raise ValueError("alltoall without splits requires dim0 divisible "
                             "by world size")
        splits_list = [rows // n] * n
    else:
        splits_list = [int(x) for x in splits]
    output = t.new_empty(0)
    received_splits = torch.empty(n, dtype=torch.int64)
    metrics.record_op('alltoall', t.numel() * t.element_size())
    core = _ensure_init()
    h = core.alltoall_async(t, output, splits_list, received_splits,
                            _auto_name("alltoall", name, psid), psid)
    return _register(h, (output, received_splits), 'alltoall')


class _DiffAlltoall(torch.autograd.Function):
    @staticmethod
    def forward(ctx, tensor, splits, name, process_set):
        out, received = synchronize(alltoall_async(tensor, splits, name,
                                                   process_set))
        ctx.args = (name, process_set)
        ctx.received = received.tolist()
        ctx.mark_non_differentiable(received)
        return out, received

    @staticmethod
    def backward(ctx, grad, _grad_received):
        name, process_set = ctx.args
        # The reverse exchange: what rank p sent us, we send back to p.
        g, _ = synchronize(alltoall_async(
            grad.contiguous(), ctx.received,
            f"{name}.grad" if name else None, process_set))
        return g, None, None, None


def alltoall(tensor, splits=None, name=None, process_set=None):
    if torch.is_grad_enabled() and tensor.requires_grad:
        out, received = _DiffAlltoall.apply(tensor, splits, name, process_set)
    else:
        out, received = synchronize(alltoall_async(tensor, splits, name,
                                                   process_set))
    if splits is None:
        return out
    return out, received


# ----------------------------------------------------------- reducescatter

def reducescatter_async(tensor, op=ReduceOp.Average, name=None,
                        prescale_factor=1.0, postscale_factor=1.0,
                        process_set=None):
    if op not in (ReduceOp.Average, ReduceOp.Sum):
        raise ValueError("reducescatter supports Average and Sum")
    psid = resolve_id(process_set)
    t = tensor.contiguous()
    output = t.new_empty(0)
    metrics.record_op('reducescatter', t.numel() * t.element_size())
    core = _ensure_init()
    h = core.reducescatter_async(t, output,
                                 _auto_name("reducescatter", name, psid),
                                 op, prescale_factor, postscale_factor, psid)
    return _register(h, output, 'reducescatter')


class _DiffReducescatter(torch.autograd.Function):
    @staticmethod
    def forward(ctx, tensor, op, name, prescale, postscale, process_set):
        ctx.args = (op, name, process_set)
        return synchronize(reducescatter_async(tensor, op, name, prescale,
                                               postscale, process_set))

    @staticmethod
    def backward(ctx, grad):
        op, name, process_set = ctx.args
        g = synchronize(allgather_async(grad.contiguous(),
                                        f"{name}.grad" if name else None,
                                        process_set))
        if op == ReduceOp.Average:
            psid = resolve_id(process_set)
            n = size() if psid == 0 else len(_ensure_init().ps_ranks(psid))
            g = g / n
        return (g,) + (None,) * 5


def reducescatter(tensor, op=ReduceOp.Average, name=None, prescale_factor=1.0,
                  postscale_factor=1.0, process_set=None):
    if torch.is_grad_enabled() and tensor.requires_grad:
        return _DiffReducescatter.apply(tensor, op, name, prescale_factor,
                                        postscale_factor, process_set)
    return synchronize(reducescatter_async(tensor, op, name, prescale_factor,
                                           postscale_factor, process_set))


# ------------------------------------------------------------ join/barrier

def join(device=-1):
    """Uneven-data barrier: a joined rank participates in
    the remaining ranks' allreduces with zero tensors until every rank has
    joined. Returns the rank of the last rank to join."""
    core = _ensure_init()
    out = torch.zeros((), dtype=torch.int64)
    h = core.join_async(device, out)
    _register(h, out)
    synchronize(h)
    return int(out.item())


def barrier(name=None):
    core = _ensure_init()
    h = core.barrier_async(_auto_name("barrier", name))
    _register(h, None)
    synchronize(h)


# ------------------------------------------------------------ handle mgmt

def synchronize(handle):
    """Block until the async op behind `handle` completes; returns its
    output tensor (or tuple)."""
    core = _ensure_init()
    core.synchronize(handle)
    with _lock:
        entry = _outputs.pop(handle, None)
    if entry is None:
        return None
    payload, kind, t0 = entry
    if kind is not None:
        metrics.record_latency(kind, time.perf_counter() - t0)
    return payload


def poll(handle):
    core = _ensure_init()
    return core.poll(handle)
