"""CPU fallback paths of the fused model ops (axon/ops/): off the GPU
fast path they must match the stock torch formulas exactly, so models are
portable and the CPU test suite stays probative."""
import torch
import torch.nn.functional as F

from axon.ops import FusedLayerNorm, FusedRMSNorm, rope, swiglu
from axon.ops.rotary import _rope_eager


def test_layernorm_cpu_fallback():
    torch.manual_seed(0)
    ref = torch.nn.LayerNorm(48)
    fused = FusedLayerNorm(48)
    fused.load_state_dict(ref.state_dict())
    x = torch.randn(7, 48, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    y1, y2 = ref(x), fused(x2)
    assert torch.allclose(y1, y2)
    y1.sum().backward()
    y2.sum().backward()
    assert torch.allclose(x.grad, x2.grad)


def test_rmsnorm_cpu_matches_reference_formula():
    torch.manual_seed(1)
    m = FusedRMSNorm(32)
    with torch.no_grad():
        m.weight.mul_(1.3)
    x = torch.randn(5, 32)
    y = m(x)
    ref = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + m.eps) * m.weight
    assert torch.allclose(y, ref, atol=1e-6)


def test_rope_cpu_preserves_norm_and_gradchecks():
    torch.manual_seed(2)
    b, s, h, d = 2, 5, 3, 8
    cos_v = torch.cos(torch.randn(s, d // 2))
    sin_v = torch.sin(torch.randn(s, d // 2))
    # proper rotation: cos^2 + sin^2 = 1 per pair -> norms preserved
    angle = torch.randn(s, d // 2)
    cos_v, sin_v = torch.cos(angle), torch.sin(angle)
    x = torch.randn(b, s, h, d)
    y = rope(x, cos_v, sin_v)
    n_in = (x[..., 0::2] ** 2 + x[..., 1::2] ** 2)
    n_out = (y[..., 0::2] ** 2 + y[..., 1::2] ** 2)
    assert torch.allclose(n_in, n_out, atol=1e-5)
    x64 = torch.randn(1, 3, 2, 4, dtype=torch.float64, requires_grad=True)
    assert torch.autograd.gradcheck(
        lambda t: _rope_eager(t, cos_v[:3, :2].double(), sin_v[:3, :2].double()),
        (x64,))


def test_swiglu_cpu_fallback():
    a = torch.randn(9, 16, requires_grad=True)
    g = torch.randn(9, 16, requires_grad=True)
    z = swiglu(a, g)
    assert torch.allclose(z, F.silu(a) * g)
    z.sum().backward()
    assert a.grad is not None and g.grad is not None
