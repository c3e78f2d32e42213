"""Fused rotary embedding + SwiGLU (csrc/kernels/elementwise.hip).

rope() runs on [B, S, H, D] BEFORE the attention transpose so the kernel
streams coalesced memory; eager torch slicing (0::2/1::2 strided views)
costs several half-efficiency kernels per call in both directions.
"""
import torch

_SUPPORTED = (torch.float32, torch.float16, torch.bfloat16)


class _RopeFunc(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cos, sin):
        from axon import _core

        ctx.save_for_backward(cos, sin)
        return _core.rope(x, cos, sin, False)

    @staticmethod
    def backward(ctx, dy):
        from axon import _core

        cos, sin = ctx.saved_tensors
        return _core.rope(dy.contiguous(), cos, sin, True), None, None


def _rope_eager(x, cos, sin):
    # x: (b, s, h, d); cos/sin: (s, d/2)
    x1, x2 = x[..., 0::2], x[..., 1::2]
    c = cos.view(1, cos.shape[0], 1, -1)
    s = sin.view(1, sin.shape[0], 1, -1)
    out = torch.empty_like(x)
    out[..., 0::2] = x1 * c - x2 * s
    out[..., 1::2] = x2 * c + x1 * s
    return out


def rope(x, cos, sin):
    """Interleaved-pair rotary embedding on (b, s, h, d)."""
    if (x.is_cuda and x.dtype in _SUPPORTED and x.is_contiguous() and
            cos.dtype == x.dtype and x.size(-1) % 2 == 0):
        return _RopeFunc.apply(x, cos.contiguous(), sin.contiguous())
    return _rope_eager(x, cos, sin)


class _SwiGLUFunc(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, g):
        from axon import _core

        ctx.save_for_backward(a, g)
        return _core.swiglu_fwd(a, g)

    @staticmethod
    def backward(ctx, dz):
        from axon import _core

        a, g = ctx.saved_tensors
        da, dg = _core.swiglu_bwd(dz, a, g)
        return da, dg


def swiglu(a, g):
    """silu(a) * g in one fused pass (backward recomputes silu in-kernel)."""
    if (a.is_cuda and a.dtype in _SUPPORTED and a.is_contiguous() and
            g.is_contiguous() and a.shape == g.shape and a.dtype == g.dtype):
        return _SwiGLUFunc.apply(a, g)
    return torch.nn.functional.silu(a) * g
