"""Fused NHWC BatchNorm2d (+ optional fused ReLU) backed by the CDNA4 HIP
kernels in csrc/kernels/batchnorm.hip.

Why: rocprof shows ATen's channels_last batch-norm kernels at ~62% of
ResNet-50 GPU time on MI355X (profiles/README.md) — they achieve a small
fraction of HBM3E bandwidth. These kernels make every pass one coalesced
stream (stats -> apply; reduce -> apply) and can fuse the trailing ReLU,
removing two more full activations passes (clamp fwd + threshold bwd).

Drop-in: subclasses nn.BatchNorm2d, identical state dict, CPU / non-NHWC /
grad-in-eval inputs fall back to the stock implementation.
"""
import torch
import torch.nn as nn
import torch.nn.functional as F

_SUPPORTED = (torch.float32, torch.float16, torch.bfloat16)


class _BatchNormFunc(torch.autograd.Function):
    """Training-mode fused BN; optionally also fuses a trailing residual
    add (bn3 + identity in a ResNet bottleneck) and/or ReLU into the same
    apply pass. `add=None` means no residual input."""

    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var, momentum,
                eps, relu, add):
        from axon import _core

        empty = torch.Tensor()
        y, save_mean, save_rstd = _core.bn_fwd(
            x, weight if weight is not None else empty,
            bias if bias is not None else empty,
            running_mean if running_mean is not None else empty,
            running_var if running_var is not None else empty,
            momentum, eps, True, relu,
            add if add is not None else empty)
        ctx.relu = relu
        ctx.has_weight = weight is not None
        ctx.has_add = add is not None
        ctx.save_for_backward(x,
                              weight if weight is not None else None,
                              save_mean, save_rstd,
                              y if relu else None)
        return y

    @staticmethod
    def backward(ctx, dy):
        from axon import _core

        x, weight, save_mean, save_rstd, y = ctx.saved_tensors
        empty = torch.Tensor()
        dx, dgamma, dbeta, dadd = _core.bn_bwd(
            x, dy, y if y is not None else empty,
            weight if weight is not None else empty,
            save_mean, save_rstd, ctx.relu, ctx.has_add)
        dw = db = None
        if ctx.has_weight:
            dw = dgamma.to(weight.dtype)
            db = dbeta.to(weight.dtype)
        if not ctx.has_add:
            dadd = None
        elif not ctx.relu:
            # without a relu mask d(add) is just dy (pass-through)
            dadd = dy
        return dx, dw, db, None, None, None, None, None, dadd


class FusedBatchNorm2d(nn.BatchNorm2d):
    """nn.BatchNorm2d with the MI355X NHWC kernels; `fuse_relu=True` folds
    the trailing ReLU into the same pass (fwd apply + bwd dy-masking)."""

    def __init__(self, num_features, eps=1e-5, momentum=0.1, affine=True,
                 track_running_stats=True, fuse_relu=False, **kw):
        super().__init__(num_features, eps=eps, momentum=momentum,
                         affine=affine,
                         track_running_stats=track_running_stats, **kw)
        self.fuse_relu = fuse_relu

    def _fast_path(self, x):
        return (x.is_cuda and x.dim() == 4 and x.dtype in _SUPPORTED and
                x.is_contiguous(memory_format=torch.channels_last) and
                x.size(1) == self.num_features)

    def _train_momentum(self):
        momentum = self.momentum if self.momentum is not None else 0.1
        if self.track_running_stats and self.num_batches_tracked is not None:
            self.num_batches_tracked.add_(1)
            if self.momentum is None:  # cumulative moving average
                momentum = 1.0 / float(self.num_batches_tracked)
        return momentum

    def forward(self, x):
        if self._fast_path(x):
            if self.training:
                rm = self.running_mean if self.track_running_stats else None
                rv = self.running_var if self.track_running_stats else None
                return _BatchNormFunc.apply(x, self.weight, self.bias, rm, rv,
                                            self._train_momentum(), self.eps,
                                            self.fuse_relu, None)
            if self.track_running_stats and not torch.is_grad_enabled():
                from axon import _core

                empty = torch.Tensor()
                y, _, _ = _core.bn_fwd(
                    x, self.weight if self.weight is not None else empty,
                    self.bias if self.bias is not None else empty,
                    self.running_mean, self.running_var, 0.0, self.eps,
                    False, self.fuse_relu, empty)
                return y
        y = super().forward(x)
        return F.relu(y, inplace=True) if self.fuse_relu else y

    def forward_add_relu(self, x, residual):
        """y = relu(bn(x) + residual) in ONE fused pass (the bottleneck
        tail). Falls back to unfused ops off the fast path."""
        if self._fast_path(x) and self.training and \
                residual.is_contiguous(memory_format=torch.channels_last):
            rm = self.running_mean if self.track_running_stats else None
            rv = self.running_var if self.track_running_stats else None
            return _BatchNormFunc.apply(x, self.weight, self.bias, rm, rv,
                                        self._train_momentum(), self.eps,
                                        True, residual)
        assert not self.fuse_relu  # bn3-style norm: relu applied here, once
        return F.relu(self.forward(x) + residual, inplace=True)
