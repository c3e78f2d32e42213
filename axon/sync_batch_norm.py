"""Cross-rank synchronized BatchNorm (hvd.SyncBatchNorm — SURVEY.md §2.2):
batch statistics are reduced across all ranks in forward, and the
statistics' gradients in backward, so DP training matches a single big
batch exactly.
"""
import itertools

import torch
import torch.nn.functional as F
from torch.nn.modules.batchnorm import _BatchNorm

from axon import mpi_ops

_instance_counter = itertools.count()


class _SyncBatchNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps, name):
        c = x.shape[1]
        dims = [0] + list(range(2, x.dim()))
        count = torch.tensor([x.numel() // c], dtype=torch.float32,
                             device=x.device)
        s = x.sum(dims).float()
        ss = (x * x).sum(dims).float()
        stats = torch.cat([s, ss, count])
        stats = mpi_ops.allreduce(stats, op=mpi_ops.Sum, name=f"{name}.fwd")
        total = stats[-1]
        mean = stats[:c] / total
        var = stats[c:2 * c] / total - mean * mean
        invstd = torch.rsqrt(var + eps)

        shape = [1, c] + [1] * (x.dim() - 2)
        xhat = (x.float() - mean.view(shape)) * invstd.view(shape)
        out = xhat * weight.float().view(shape) + bias.float().view(shape)
        ctx.save_for_backward(x, weight, mean, invstd, total)
        ctx.name = name
        ctx.dims = dims
        return out.to(x.dtype), mean, var

    @staticmethod
    def backward(ctx, grad_out, _gm, _gv):
        x, weight, mean, invstd, total = ctx.saved_tensors
        c = x.shape[1]
        dims = ctx.dims
        shape = [1, c] + [1] * (x.dim() - 2)
        g = grad_out.float()
        xhat = (x.float() - mean.view(shape)) * invstd.view(shape)

        sum_g = g.sum(dims)
        sum_gx = (g * xhat).sum(dims)
        red = mpi_ops.allreduce(torch.cat([sum_g, sum_gx]), op=mpi_ops.Sum,
                                name=f"{ctx.name}.bwd")
        sum_g, sum_gx = red[:c], red[c:]

        grad_weight = (g * xhat).sum(dims).to(weight.dtype)
        grad_bias = g.sum(dims).to(weight.dtype)
        gi = (g - (sum_g / total).view(shape)
              - xhat * (sum_gx / total).view(shape)) \
            * invstd.view(shape) * weight.float().view(shape)
        return gi.to(x.dtype), grad_weight, grad_bias, None, None


class SyncBatchNorm(_BatchNorm):
    """Drop-in BatchNorm whose statistics span all ranks. Falls back to
    regular batch norm when world_size == 1 or in eval mode."""

    def __init__(self, num_features, eps=1e-5, momentum=0.1, affine=True,
                 track_running_stats=True):
        super().__init__(num_features, eps, momentum, affine,
                         track_running_stats)
        self._name = f"syncbn.{next(_instance_counter)}"
        self._call = 0

    def _check_input_dim(self, x):
        if x.dim() < 2:
            raise ValueError("SyncBatchNorm expects at least 2D input")

    def forward(self, x):
        self._check_input_dim(x)
        if not self.training or mpi_ops.size() == 1:
            return F.batch_norm(x, self.running_mean, self.running_var,
                                self.weight, self.bias, self.training,
                                self.momentum, self.eps)
        name = f"{self._name}.{self._call}"
        self._call += 1
        out, mean, var = _SyncBatchNormFn.apply(x, self.weight, self.bias,
                                                self.eps, name)
        if self.track_running_stats:
            with torch.no_grad():
                m = self.momentum if self.momentum is not None else 0.1
                n = x.numel() // x.shape[1] * mpi_ops.size()
                unbiased = var * n / max(n - 1, 1)
                self.running_mean.mul_(1 - m).add_(m * mean.to(
                    self.running_mean.dtype))
                self.running_var.mul_(1 - m).add_(m * unbiased.to(
                    self.running_var.dtype))
                self.num_batches_tracked += 1
        return out

    @classmethod
    def convert_sync_batchnorm(cls, module):
        """Recursively replace BatchNorm layers with SyncBatchNorm."""
        out = module
        if isinstance(module, _BatchNorm) and not isinstance(module, cls):
            out = cls(module.num_features, module.eps, module.momentum,
                      module.affine, module.track_running_stats)
            if module.affine:
                with torch.no_grad():
                    out.weight.copy_(module.weight)
                    out.bias.copy_(module.bias)
            out.running_mean = module.running_mean
            out.running_var = module.running_var
            out.num_batches_tracked = module.num_batches_tracked
            out = out.to(module.weight.device if module.affine else "cpu")
        for name, child in module.named_children():
            out.add_module(name, cls.convert_sync_batchnorm(child))
        return out
