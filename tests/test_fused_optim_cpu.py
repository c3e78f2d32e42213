"""Fused optimizer semantics (CPU fallback path) vs stock torch optimizers
— the same formulas the HIP kernels implement (GPU parity is asserted in
tests/test_gpu.py against this oracle).
"""
import math

import pytest
import torch
import torch.nn as nn

from axon.optim import FusedAdam, FusedAdamW, FusedLAMB, FusedSGD


def _pair(seed=0):
    torch.manual_seed(seed)
    m1 = nn.Sequential(nn.Linear(7, 9), nn.Tanh(), nn.Linear(9, 4))
    torch.manual_seed(seed)
    m2 = nn.Sequential(nn.Linear(7, 9), nn.Tanh(), nn.Linear(9, 4))
    return m1, m2


def _train(model, opt, steps=5, seed=11):
    for s in range(steps):
        torch.manual_seed(seed + s)
        x = torch.randn(6, 7)
        loss = (model(x) ** 2).mean()
        opt.zero_grad()
        loss.backward()
        opt.step()


@pytest.mark.parametrize("momentum,nesterov,wd", [
    (0.0, False, 0.0), (0.9, False, 0.0), (0.9, True, 0.0), (0.9, False, 1e-2),
])
def test_fused_sgd_matches_torch(momentum, nesterov, wd):
    m1, m2 = _pair()
    o1 = FusedSGD(m1.parameters(), lr=0.05, momentum=momentum,
                  nesterov=nesterov, weight_decay=wd)
    o2 = torch.optim.SGD(m2.parameters(), lr=0.05, momentum=momentum,
                         nesterov=nesterov, weight_decay=wd)
    _train(m1, o1)
    _train(m2, o2)
    for p, q in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p, q, rtol=1e-5, atol=1e-6)


@pytest.mark.parametrize("cls,ref,wd", [
    (FusedAdam, torch.optim.Adam, 0.0),
    (FusedAdam, torch.optim.Adam, 1e-2),
    (FusedAdamW, torch.optim.AdamW, 1e-2),
])
def test_fused_adam_matches_torch(cls, ref, wd):
    m1, m2 = _pair(3)
    o1 = cls(m1.parameters(), lr=2e-3, weight_decay=wd)
    o2 = ref(m2.parameters(), lr=2e-3, weight_decay=wd)
    _train(m1, o1, steps=6)
    _train(m2, o2, steps=6)
    for p, q in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p, q, rtol=1e-4, atol=1e-6), (p - q).abs().max()


def test_fused_lamb_reference():
    """LAMB oracle: hand-computed trust-ratio update on a single tensor."""
    torch.manual_seed(5)
    p = torch.randn(64)
    g = torch.randn(64)
    model_p = p.clone().requires_grad_(True)
    opt = FusedLAMB([model_p], lr=0.01, weight_decay=0.02)
    model_p.grad = g.clone()
    opt.step()

    b1, b2, eps = 0.9, 0.999, 1e-6
    m = (1 - b1) * g
    v = (1 - b2) * g * g
    mh = m / (1 - b1)
    vh = v.sqrt() / math.sqrt(1 - b2)
    u = mh / (vh + eps) + 0.02 * p
    ratio = p.norm() / u.norm()
    expect = p - 0.01 * ratio * u
    assert torch.allclose(model_p.detach(), expect, rtol=1e-5, atol=1e-6)


def test_fused_state_dict_interchange():
    """Fused optimizer state loads into the stock torch optimizer."""
    m1, m2 = _pair(9)
    o1 = FusedAdam(m1.parameters(), lr=1e-3)
    _train(m1, o1, steps=3)
    sd = o1.state_dict()
    o2 = torch.optim.Adam(m2.parameters(), lr=1e-3)
    o2.load_state_dict(sd)  # must not raise
    _train(m2, o2, steps=1)
    st = list(o2.state.values())[0]
    assert "exp_avg" in st and "exp_avg_sq" in st
    assert float(st["step"]) == 4


# ------------------------- property sweeps (hypothesis) -------------------

from hypothesis import given, settings, strategies as st  # noqa: E402


@settings(max_examples=25, deadline=None, derandomize=True)
@given(lr=st.floats(1e-4, 0.5),
       momentum=st.one_of(st.just(0.0), st.floats(1e-3, 0.99)),
       dampening=st.floats(0.0, 0.9), wd=st.floats(0.0, 0.05),
       maximize=st.booleans(), seed=st.integers(0, 10_000))
def test_fused_sgd_property_sweep(lr, momentum, dampening, wd, maximize,
                                  seed):
    m1, m2 = _pair(seed % 100)
    kw = dict(lr=lr, momentum=momentum, dampening=dampening,
              weight_decay=wd, maximize=maximize)
    _train(m1, FusedSGD(m1.parameters(), **kw), steps=3, seed=seed)
    _train(m2, torch.optim.SGD(m2.parameters(), **kw), steps=3, seed=seed)
    for p, q in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p, q, rtol=1e-4, atol=1e-6), kw


@settings(max_examples=20, deadline=None, derandomize=True)
@given(lr=st.floats(1e-5, 1e-2), b1=st.floats(0.5, 0.95),
       b2=st.floats(0.9, 0.9999), eps=st.floats(1e-10, 1e-6),
       wd=st.floats(0.0, 0.05), adamw=st.booleans(),
       seed=st.integers(0, 10_000))
def test_fused_adam_property_sweep(lr, b1, b2, eps, wd, adamw, seed):
    m1, m2 = _pair(seed % 100)
    kw = dict(lr=lr, betas=(b1, b2), eps=eps, weight_decay=wd)
    fused = (FusedAdamW if adamw else FusedAdam)(m1.parameters(), **kw)
    ref = (torch.optim.AdamW if adamw else torch.optim.Adam)(
        m2.parameters(), **kw)
    _train(m1, fused, steps=3, seed=seed)
    _train(m2, ref, steps=3, seed=seed)
    for p, q in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p, q, rtol=1e-4, atol=1e-6), kw
