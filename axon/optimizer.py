"""DistributedOptimizer: backward/communication overlap via per-parameter
autograd hooks.

Capability parity: horovod/torch/optimizer.py (SURVEY.md §2.2, §3.2 "the
money path"): each parameter gets a post-accumulate-grad hook that fires an
in-place async allreduce the moment the gradient is ready; step() first
synchronizes all outstanding handles. The background core overlaps the
fused RCCL allreduces with the rest of backward on a dedicated HIP stream.

Implementation note: like the reference, the wrapped optimizer keeps its
own class in the MRO (so isinstance checks and LR schedulers keep working);
we re-class the instance with a mixin that overrides step/zero_grad.
"""
from contextlib import contextmanager

from axon import mpi_ops
from axon.compression import Compression
from axon.mpi_ops import ReduceOp


class _DistributedOptimizer:
    def _distributed_init(self, named_parameters, compression,
                          backward_passes_per_step, op,
                          gradient_predivide_factor, process_set,
                          sparse_as_dense=False):
        self._sparse_as_dense = sparse_as_dense
        self._compression = compression
        self._bpps = backward_passes_per_step
        self._op = op
        self._predivide = gradient_predivide_factor
        self._process_set = process_set
        self._ps_size = (mpi_ops.size() if process_set is None
                         else len(process_set.ranks))

        if named_parameters is not None:
            name_of = {id(p): n for n, p in list(named_parameters)}
        else:
            name_of = {}
            for gi, group in enumerate(self.param_groups):
                for pi, p in enumerate(group["params"]):
                    name_of[id(p)] = f"group{gi}.param{pi}"

        self._dist_params = [p for group in self.param_groups
                             for p in group["params"] if p.requires_grad]
        self._names = {id(p): name_of.get(id(p), f"param{i}")
                       for i, p in enumerate(self._dist_params)}
        self._handles = {}
        self._passes = {id(p): 0 for p in self._dist_params}
        self._hook_handles = []
        self._require_sync = True
        if self._ps_size > 1 and (self._process_set is None or
                                  self._process_set.included()):
            self._register_hooks()

    def _register_hooks(self):
        for p in self._dist_params:
            self._hook_handles.append(
                p.register_post_accumulate_grad_hook(self._make_hook(p)))

    def _make_hook(self, p):
        def hook(_param):
            self._passes[id(p)] += 1
            if self._passes[id(p)] >= self._bpps:
                self._passes[id(p)] = 0
                self._allreduce_grad_async(p)
        return hook

    def _allreduce_grad_async(self, p):
        if p.grad is None:
            return
        if p.grad.is_sparse:
            # Sparse grads (e.g. Embedding(sparse=True)): the collective
            # wire is dense; hvd-parity sparse_as_dense densifies before
            # the allreduce, otherwise fail with the actionable fix.
            if not self._sparse_as_dense:
                raise ValueError(
                    f"parameter '{self._names[id(p)]}' has a sparse "
                    "gradient; pass sparse_as_dense=True to "
                    "DistributedOptimizer (or use a dense embedding)")
            p.grad = p.grad.to_dense()
        name = f"grad.{self._names[id(p)]}"
        if self._predivide != 1.0:
            handle = mpi_ops.allreduce_async_(
                p.grad, op=ReduceOp.Sum, name=name,
                prescale_factor=1.0 / self._predivide,
                postscale_factor=self._predivide / self._ps_size,
                compression=self._compression,
                process_set=self._process_set)
        else:
            handle = mpi_ops.allreduce_async_(
                p.grad, op=self._op, name=name,
                compression=self._compression,
                process_set=self._process_set)
        self._handles[id(p)] = handle

    def set_backward_passes_per_step(self, passes):
        """Change gradient-accumulation depth (hvd API): allreduce fires on
        every `passes`-th backward for each parameter."""
        self._bpps = passes
        for p in self._dist_params:
            self._passes[id(p)] = 0

    def synchronize(self):
        """Wait for every outstanding gradient allreduce."""
        for h in self._handles.values():
            mpi_ops.synchronize(h)
        self._handles.clear()

    @contextmanager
    def skip_synchronize(self):
        """Run step() without the implicit synchronize (the caller already
        synchronized — e.g. after gradient clipping)."""
        prev = self._require_sync
        self._require_sync = False
        try:
            yield
        finally:
            self._require_sync = prev

    def step(self, closure=None):
        if self._require_sync and self._ps_size > 1:
            self.synchronize()
        ret = super().step(closure)
        # End-to-end autotune signal: steps/sec on rank 0 scores the
        # fusion-threshold/cycle-time sweep (no-op unless AXON_AUTOTUNE=1).
        from axon import _core, metrics
        _core.autotune_step_mark()
        metrics.record_step()
        return ret

    def zero_grad(self, set_to_none=True):
        if self._handles:
            raise RuntimeError(
                "zero_grad called with allreduce handles outstanding — call "
                "optimizer.step() or synchronize() first")
        return super().zero_grad(set_to_none=set_to_none)


def DistributedOptimizer(optimizer, named_parameters=None,
                         compression=Compression.none,
                         backward_passes_per_step=1, op=ReduceOp.Average,
                         gradient_predivide_factor=1.0, process_set=None,
                         sparse_as_dense=False):
    """Wrap a torch optimizer with gradient averaging over all ranks — or
    over a registered ProcessSet's members (hvd.DistributedOptimizer API).
    sparse_as_dense densifies sparse gradients (Embedding(sparse=True))
    before the allreduce, matching the reference's option."""
    cls = type(optimizer.__class__.__name__, (_DistributedOptimizer,
                                              optimizer.__class__), {})
    optimizer.__class__ = cls
    optimizer._distributed_init(named_parameters, compression,
                                backward_passes_per_step, op,
                                gradient_predivide_factor, process_set,
                                sparse_as_dense)
    return optimizer
