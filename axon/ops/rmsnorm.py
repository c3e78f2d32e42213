"""Fused RMSNorm module backed by csrc/kernels/rmsnorm.hip (one kernel and
one global pass per direction vs ~7 eager kernels; see profiles/README.md
llama notes). Falls back to the eager fp32 formula off the fast path."""
import torch
import torch.nn as nn

_SUPPORTED = (torch.float32, torch.float16, torch.bfloat16)
_SUPPORTED_DIMS = {64, 128, 256, 512, 1024, 2048, 4096}


class _RMSNormFunc(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        from axon import _core

        y, rstd = _core.rms_fwd(x, weight, eps)
        ctx.save_for_backward(x, weight, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        from axon import _core

        x, weight, rstd = ctx.saved_tensors
        dx, dw = _core.rms_bwd(x, dy, weight, rstd)
        return dx, dw.to(weight.dtype), None


class FusedRMSNorm(nn.Module):
    def __init__(self, dim, eps=1e-5):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.eps = eps
        self.dim = dim

    def _fast_path(self, x):
        return (x.is_cuda and x.dtype in _SUPPORTED and
                x.dtype == self.weight.dtype and x.size(-1) == self.dim and
                self.dim in _SUPPORTED_DIMS and x.is_contiguous())

    def forward(self, x):
        if self._fast_path(x):
            return _RMSNormFunc.apply(x, self.weight, self.eps)
        dt = x.dtype
        xf = x.float()
        xf = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + self.eps)
        return (xf * self.weight.float()).to(dt)
