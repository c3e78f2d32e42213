"""Fused optimizer wrappers — see axon/optim/__init__.py docstring."""
import math

import torch
from torch.optim import Optimizer


def _like_param(p, t):
    """The fused kernels walk flat storage, so grad/state must share the
    param's strides (channels_last params included). Autograd or a loaded
    checkpoint can hand us a differently-laid-out tensor — restride it once.
    Grads are read-only in the kernels, so a copy is safe."""
    if t is None or p.numel() <= 1 or t.stride() == p.stride():
        return t
    out = torch.empty_strided(p.shape, p.stride(), dtype=t.dtype,
                              device=t.device)
    out.copy_(t)
    return out


def _grouped(params_with_grad):
    """Group (param, grad, state...) lists by (device, dtype)."""
    groups = {}
    for tup in params_with_grad:
        p = tup[0]
        groups.setdefault((p.device, p.dtype), []).append(tup)
    return groups.values()


class FusedSGD(Optimizer):
    """torch.optim.SGD semantics, one HIP kernel launch per group."""

    def __init__(self, params, lr=0.1, momentum=0.0, dampening=0.0,
                 weight_decay=0.0, nesterov=False, maximize=False):
        if nesterov and (momentum <= 0 or dampening != 0):
            raise ValueError("nesterov requires momentum > 0, dampening 0")
        if dampening != 0 and 0 < momentum < 1e-8:
            # The fused kernel's uniform m = mu*m + (1-damp)*d update
            # reproduces torch's undamped first step via a (damp/mu)-scaled
            # seed buffer, which overflows for denormal momentum. No real
            # configuration pairs ~zero momentum with dampening.
            raise ValueError("fused SGD: momentum in (0, 1e-8) with "
                             "nonzero dampening is unsupported")
        super().__init__(params, dict(lr=lr, momentum=momentum,
                                      dampening=dampening,
                                      weight_decay=weight_decay,
                                      nesterov=nesterov, maximize=maximize))

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        from axon import _core

        for group in self.param_groups:
            work = []
            mu = group["momentum"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                st = self.state[p]
                if mu != 0 and "momentum_buffer" not in st:
                    damp = group["dampening"]
                    if damp == 0:
                        st["momentum_buffer"] = torch.zeros_like(p)
                    else:
                        # torch.optim.SGD seeds the buffer with the raw
                        # first-step update direction (dampening is NOT
                        # applied on init). A zero buffer only reproduces
                        # that when dampening == 0; seeding with
                        # (damp/mu)*d0 makes the uniform kernel update
                        # mu*m + (1-damp)*d equal d on the first step.
                        g0 = _like_param(p, p.grad)
                        d0 = (-g0 if group["maximize"] else g0).float()
                        if group["weight_decay"] != 0:
                            d0 = d0 + group["weight_decay"] * p.detach().float()
                        st["momentum_buffer"] = \
                            (d0 * (damp / mu)).to(p.dtype)
                elif mu != 0:
                    st["momentum_buffer"] = _like_param(p, st["momentum_buffer"])
                work.append((p, _like_param(p, p.grad),
                             st["momentum_buffer"] if mu != 0 else None))
            for tuples in _grouped(work):
                ps = [t[0] for t in tuples]
                gs = [t[1] for t in tuples]
                ms = [t[2] for t in tuples] if mu != 0 else []
                if ps[0].is_cuda:
                    _core.fused_sgd([p.data for p in ps], gs, ms, group["lr"],
                                    mu, group["dampening"],
                                    group["weight_decay"], group["nesterov"],
                                    group["maximize"])
                else:
                    self._step_cpu(ps, gs, ms, group)
        return loss

    @staticmethod
    def _step_cpu(ps, gs, ms, group):
        for i, p in enumerate(ps):
            g = gs[i].float()
            if group["maximize"]:
                g = -g
            d = g + group["weight_decay"] * p.float()
            if group["momentum"] != 0:
                m = ms[i]
                mf = group["momentum"] * m.float() + (1 - group["dampening"]) * d
                m.copy_(mf)
                d = d + group["momentum"] * mf if group["nesterov"] else mf
            p.data.copy_(p.float() - group["lr"] * d)


class _FusedAdamBase(Optimizer):
    _adamw = False

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.0):
        super().__init__(params, dict(lr=lr, betas=betas, eps=eps,
                                      weight_decay=weight_decay))

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        from axon import _core

        for group in self.param_groups:
            work = []
            for p in group["params"]:
                if p.grad is None:
                    continue
                st = self.state[p]
                if "exp_avg" not in st:
                    st["step"] = torch.tensor(0.0)
                    st["exp_avg"] = torch.zeros_like(p)
                    st["exp_avg_sq"] = torch.zeros_like(p)
                st["step"] += 1
                st["exp_avg"] = _like_param(p, st["exp_avg"])
                st["exp_avg_sq"] = _like_param(p, st["exp_avg_sq"])
                work.append((p, _like_param(p, p.grad), st["exp_avg"],
                             st["exp_avg_sq"], int(st["step"].item())))
            b1, b2 = group["betas"]
            for tuples in _grouped(work):
                # all params in a (device,dtype) bucket share the step count
                # only if they were created together; split by step to be safe
                by_step = {}
                for t in tuples:
                    by_step.setdefault(t[4], []).append(t)
                for step_count, tl in by_step.items():
                    ps = [t[0] for t in tl]
                    gs = [t[1] for t in tl]
                    ms = [t[2] for t in tl]
                    vs = [t[3] for t in tl]
                    if ps[0].is_cuda:
                        _core.fused_adam([p.data for p in ps], gs, ms, vs,
                                         group["lr"], b1, b2, group["eps"],
                                         group["weight_decay"], step_count,
                                         self._adamw)
                    else:
                        self._step_cpu(ps, gs, ms, vs, group, step_count,
                                       self._adamw)
        return loss

    @staticmethod
    def _step_cpu(ps, gs, ms, vs, group, step, adamw):
        b1, b2 = group["betas"]
        wd = group["weight_decay"]
        bc1 = 1 - b1 ** step
        bc2s = math.sqrt(1 - b2 ** step)
        for i, p in enumerate(ps):
            g = gs[i].float()
            pf = p.float()
            if adamw:
                pf = pf * (1 - group["lr"] * wd)
            else:
                g = g + wd * pf
            mf = b1 * ms[i].float() + (1 - b1) * g
            vf = b2 * vs[i].float() + (1 - b2) * g * g
            ms[i].copy_(mf)
            vs[i].copy_(vf)
            mh = mf / bc1
            vh = vf.sqrt() / bc2s
            p.data.copy_(pf - group["lr"] * mh / (vh + group["eps"]))


class FusedAdam(_FusedAdamBase):
    """torch.optim.Adam semantics (L2-style weight decay into the grad)."""
    _adamw = False


class FusedAdamW(_FusedAdamBase):
    """torch.optim.AdamW semantics (decoupled weight decay)."""
    _adamw = True


class FusedLAMB(Optimizer):
    """LAMB (You et al., Large Batch Optimization for Deep Learning):
    Adam-style update direction with decoupled weight decay, scaled per
    tensor by trust ratio ||p|| / ||update||. Two fused HIP phases: phase 1
    computes updates + per-tensor norms (MFMA lane reduction + LDS tree),
    phase 2 applies the trust-ratio step. BERT-large pretrain config
    (BASELINE.json config 3) uses this."""

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-6,
                 weight_decay=0.0):
        super().__init__(params, dict(lr=lr, betas=betas, eps=eps,
                                      weight_decay=weight_decay))

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        from axon import _core

        for group in self.param_groups:
            work = []
            for p in group["params"]:
                if p.grad is None:
                    continue
                st = self.state[p]
                if "exp_avg" not in st:
                    st["step"] = torch.tensor(0.0)
                    st["exp_avg"] = torch.zeros_like(p)
                    st["exp_avg_sq"] = torch.zeros_like(p)
                st["step"] += 1
                st["exp_avg"] = _like_param(p, st["exp_avg"])
                st["exp_avg_sq"] = _like_param(p, st["exp_avg_sq"])
                work.append((p, _like_param(p, p.grad), st["exp_avg"],
                             st["exp_avg_sq"], int(st["step"].item())))
            b1, b2 = group["betas"]
            for tuples in _grouped(work):
                by_step = {}
                for t in tuples:
                    by_step.setdefault(t[4], []).append(t)
                for step_count, tl in by_step.items():
                    ps = [t[0] for t in tl]
                    gs = [t[1] for t in tl]
                    ms = [t[2] for t in tl]
                    vs = [t[3] for t in tl]
                    if ps[0].is_cuda:
                        _core.fused_lamb([p.data for p in ps], gs, ms, vs,
                                         group["lr"], b1, b2, group["eps"],
                                         group["weight_decay"], step_count)
                    else:
                        self._step_cpu(ps, gs, ms, vs, group, step_count)
        return loss

    @staticmethod
    def _step_cpu(ps, gs, ms, vs, group, step):
        b1, b2 = group["betas"]
        bc1 = 1 - b1 ** step
        bc2s = math.sqrt(1 - b2 ** step)
        for i, p in enumerate(ps):
            g = gs[i].float()
            pf = p.float()
            mf = b1 * ms[i].float() + (1 - b1) * g
            vf = b2 * vs[i].float() + (1 - b2) * g * g
            ms[i].copy_(mf)
            vs[i].copy_(vf)
            u = (mf / bc1) / (vf.sqrt() / bc2s + group["eps"]) \
                + group["weight_decay"] * pf
            pn = pf.norm()
            un = u.norm()
            ratio = (pn / un).item() if pn > 0 and un > 0 else 1.0
            p.data.copy_(pf - group["lr"] * ratio * u)
