"""CDNA4 flash attention (axon/csrc/kernels/flash_attn.hip) with an SDPA
fallback. Forward: 8-wave mfma_32x32x16 online-softmax kernel (swapped
QK^T, in-register softmax, transposed-V LDS image); backward: FA-2-style
recompute from the saved logsumexp (delta + dkdv + dq kernels, GQA
accumulated inside the k-tile wave — no atomics, no workspace).

Enable in the models with AXON_FA=1 (opt-in until the bench deltas are
in; the fallback path is torch SDPA, so the switch is always safe).

Capability parity note: the reference stack has no
attention kernels of its own — this is a beyond-parity MI355X-native op.
"""
import os

import torch
import torch.nn.functional as F


def _core():
    from axon import _core as c
    return c


def _supported(q, k, v, causal):
    if not (q.is_cuda and causal):
        return False
    if q.dtype != torch.bfloat16:
        return False
    if q.shape[-1] != 128:
        return False
    hq, hk = q.shape[1], k.shape[1]
    if hq % hk != 0:
        return False
    return True


class _FlashAttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v):
        q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        o, lse = _core().fa_fwd(q, k, v, True)
        ctx.save_for_backward(q, k, v, o, lse)
        return o

    @staticmethod
    def backward(ctx, grad):
        q, k, v, o, lse = ctx.saved_tensors
        dq, dk, dv = _core().fa_bwd(q, k, v, o, grad.contiguous(), lse, True)
        return dq, dk, dv


def flash_attention(q, k, v, causal=True):
    """q: (b, hq, s, d), k/v: (b, hk, s, d) bf16 CUDA tensors; causal.
    Routes through the CDNA4 kernels when supported, torch SDPA otherwise.
    AXON_FA=0 forces the SDPA path."""
    if os.environ.get("AXON_FA", "0") == "1" and _supported(q, k, v, causal):
        return _FlashAttnFn.apply(q, k, v)
    return F.scaled_dot_product_attention(q, k, v, is_causal=causal,
                                          enable_gqa=True)
