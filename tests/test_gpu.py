"""GPU (MI355X) tests — run via `pytest -m gpu` on a GPU box (gpurun).
Single-GPU world: RCCL 1-rank communicator + the CDNA4 kernels. Kernel
numerics are asserted against the plain fp32 Python oracles
(tests/test_fused_optim_cpu.py formulas / torch casts).
"""
import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs a GPU")


@pytest.fixture(scope="module")
def hvd():
    import axon as hvd

    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    hvd.init()
    yield hvd
    hvd.shutdown()


DEV = "cuda:0"


# ------------------------------------------------- engine collectives ws=1

def test_allreduce_direct(hvd):
    t = torch.randn(1 << 20, device=DEV)
    out = hvd.allreduce(t, average=True)
    assert torch.allclose(out, t)


def test_allreduce_fused_pack_unpack(hvd):
    ts = [torch.randn(1000 + 37 * i, device=DEV) for i in range(30)]
    outs = hvd.grouped_allreduce(ts, op=hvd.Sum, name="gpu_fused")
    for t, o in zip(ts, outs):
        assert torch.allclose(o, t, rtol=1e-6, atol=1e-6)


def test_allreduce_scales_gpu(hvd):
    t = torch.full((12345,), 2.0, device=DEV)
    out = hvd.allreduce(t, op=hvd.Sum, prescale_factor=0.5,
                        postscale_factor=3.0, name="gpu_scaled")
    assert torch.allclose(out, torch.full_like(t, 3.0))


def test_fp16_wire_kernels(hvd):
    ts = [torch.randn(777 + i, device=DEV) * 10 for i in range(5)]
    outs = hvd.grouped_allreduce(ts, op=hvd.Sum, name="gpu_fp16",
                                 compression=hvd.Compression.fp16)
    for t, o in zip(ts, outs):
        ref = t.half().float()  # fp32 torch oracle of the wire cast
        assert o.dtype == torch.float32
        assert torch.allclose(o, ref, rtol=1e-3, atol=1e-3)


def test_bf16_fused(hvd):
    ts = [torch.randn(513 + i, device=DEV).bfloat16() for i in range(8)]
    outs = hvd.grouped_allreduce(ts, op=hvd.Sum, name="gpu_bf16")
    for t, o in zip(ts, outs):
        assert o.dtype == torch.bfloat16
        assert torch.allclose(o.float(), t.float())


def test_broadcast_gpu(hvd):
    t = torch.randn(4096, device=DEV)
    out = hvd.broadcast(t, root_rank=0)
    assert torch.allclose(out, t)


def test_allgather_gpu(hvd):
    t = torch.randn(33, 5, device=DEV)
    out = hvd.allgather(t)
    assert out.shape == (33, 5)
    assert torch.allclose(out, t)


def test_alltoall_gpu(hvd):
    t = torch.randn(8, 3, device=DEV)
    out = hvd.alltoall(t)
    assert torch.allclose(out, t)


def test_reducescatter_gpu(hvd):
    t = torch.randn(10, 4, device=DEV)
    out = hvd.reducescatter(t, op=hvd.Sum)
    assert torch.allclose(out, t)  # ws=1: identity


def test_adasum_gpu_ws1(hvd):
    t = torch.randn(999, device=DEV)
    out = hvd.allreduce(t, op=hvd.Adasum, name="gpu_adasum")
    assert torch.allclose(out, t)


def test_large_single_tensor(hvd):
    # > fusion threshold: exercises the scratch/direct path
    t = torch.randn(40 << 20, device=DEV)  # 160 MB fp32
    out = hvd.allreduce(t, op=hvd.Sum, name="gpu_big")
    assert torch.allclose(out, t)


# ------------------------------------------------------ fused optimizers

def _models(dtype, seed=0):
    import torch.nn as nn

    torch.manual_seed(seed)
    gpu = nn.Sequential(nn.Linear(64, 128), nn.GELU(),
                        nn.Linear(128, 32)).to(DEV, dtype)
    cpu = nn.Sequential(nn.Linear(64, 128), nn.GELU(), nn.Linear(128, 32))
    cpu.load_state_dict({k: v.float().cpu() for k, v in gpu.state_dict().items()})
    return gpu, cpu


def _run_pair(gpu_model, cpu_model, gpu_opt, cpu_opt, dtype, steps=5):
    for s in range(steps):
        torch.manual_seed(100 + s)
        x = torch.randn(16, 64)
        xg = x.to(DEV, dtype)
        (gpu_model(xg).float() ** 2).mean().backward()
        (cpu_model(x) ** 2).mean().backward()
        gpu_opt.step()
        cpu_opt.step()
        gpu_opt.zero_grad()
        cpu_opt.zero_grad()


@pytest.mark.parametrize("dtype,tol", [(torch.float32, 1e-4),
                                       (torch.bfloat16, 5e-2)])
def test_fused_sgd_gpu(hvd, dtype, tol):
    from axon.optim import FusedSGD

    gpu_m, cpu_m = _models(dtype, seed=1)
    go = FusedSGD(gpu_m.parameters(), lr=0.05, momentum=0.9, weight_decay=1e-3)
    co = FusedSGD(cpu_m.parameters(), lr=0.05, momentum=0.9, weight_decay=1e-3)
    _run_pair(gpu_m, cpu_m, go, co, dtype)
    for p, q in zip(gpu_m.parameters(), cpu_m.parameters()):
        assert torch.allclose(p.float().cpu(), q.detach(), rtol=tol, atol=tol), \
            (p.float().cpu() - q).abs().max()


@pytest.mark.parametrize("cls", ["adam", "adamw", "lamb"])
def test_fused_adam_family_gpu(hvd, cls):
    from axon.optim import FusedAdam, FusedAdamW, FusedLAMB

    ctor = {"adam": FusedAdam, "adamw": FusedAdamW, "lamb": FusedLAMB}[cls]
    gpu_m, cpu_m = _models(torch.float32, seed=2)
    go = ctor(gpu_m.parameters(), lr=2e-3, weight_decay=1e-2)
    co = ctor(cpu_m.parameters(), lr=2e-3, weight_decay=1e-2)
    _run_pair(gpu_m, cpu_m, go, co, torch.float32)
    for p, q in zip(gpu_m.parameters(), cpu_m.parameters()):
        assert torch.allclose(p.float().cpu(), q.detach(), rtol=1e-3, atol=1e-4), \
            (p.float().cpu() - q).abs().max()


def test_fused_sgd_matches_torch_gpu(hvd):
    """Fused HIP SGD vs stock torch SGD on the SAME GPU data."""
    import torch.nn as nn

    from axon.optim import FusedSGD

    torch.manual_seed(7)
    m1 = nn.Linear(256, 256).to(DEV)
    m2 = nn.Linear(256, 256).to(DEV)
    m2.load_state_dict(m1.state_dict())
    o1 = FusedSGD(m1.parameters(), lr=0.1, momentum=0.9)
    o2 = torch.optim.SGD(m2.parameters(), lr=0.1, momentum=0.9)
    for s in range(4):
        torch.manual_seed(s)
        x = torch.randn(32, 256, device=DEV)
        (m1(x) ** 2).mean().backward()
        (m2(x) ** 2).mean().backward()
        o1.step()
        o2.step()
        o1.zero_grad()
        o2.zero_grad()
    for p, q in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p, q, rtol=1e-5, atol=1e-6)


@pytest.mark.parametrize("dtype,tol", [(torch.float32, 2e-4),
                                       (torch.bfloat16, 3e-2)])
@pytest.mark.parametrize("C,HW", [(64, 14), (128, 9), (256, 7), (96, 5),
                                  (512, 4)])
def test_fused_batchnorm_matches_torch(hvd, dtype, tol, C, HW):
    """Fused NHWC BN (kernels/batchnorm.hip) vs stock nn.BatchNorm2d:
    fwd output, input/weight/bias grads, and running stats."""
    from axon.ops import FusedBatchNorm2d

    torch.manual_seed(C)
    ref = torch.nn.BatchNorm2d(C).to(DEV)
    fused = FusedBatchNorm2d(C).to(DEV)
    fused.load_state_dict(ref.state_dict())
    if dtype != torch.float32:
        ref = ref.to(dtype)
        fused = fused.to(dtype)
    x = torch.randn(8, C, HW, HW, device=DEV, dtype=dtype) * 3 + 1
    x1 = x.clone().requires_grad_(True)
    x2 = x.clone().to(memory_format=torch.channels_last).requires_grad_(True)
    y1 = ref(x1)
    y2 = fused(x2)
    assert torch.allclose(y1, y2.to(y1.dtype), rtol=tol, atol=tol), \
        (y1 - y2).abs().max()
    g = torch.randn_like(y1)
    y1.backward(g)
    y2.backward(g.to(memory_format=torch.channels_last))
    assert torch.allclose(x1.grad, x2.grad, rtol=tol, atol=tol), \
        (x1.grad - x2.grad).abs().max()
    assert torch.allclose(ref.weight.grad, fused.weight.grad,
                          rtol=tol, atol=tol * 10)
    assert torch.allclose(ref.bias.grad, fused.bias.grad,
                          rtol=tol, atol=tol * 10)
    assert torch.allclose(ref.running_mean.float(),
                          fused.running_mean.float(), rtol=tol, atol=tol)
    assert torch.allclose(ref.running_var.float(),
                          fused.running_var.float(), rtol=tol, atol=tol)
    # eval mode (inference path, no grad)
    ref.eval()
    fused.eval()
    with torch.no_grad():
        ye1 = ref(x)
        ye2 = fused(x.to(memory_format=torch.channels_last))
    assert torch.allclose(ye1, ye2.to(ye1.dtype), rtol=tol, atol=tol)


def test_fused_batchnorm_relu(hvd):
    """fuse_relu folds the activation into fwd apply and masks dy in bwd."""
    from axon.ops import FusedBatchNorm2d

    torch.manual_seed(3)
    C = 64
    ref = torch.nn.BatchNorm2d(C).to(DEV)
    fused = FusedBatchNorm2d(C, fuse_relu=True).to(DEV)
    fused.load_state_dict(ref.state_dict())
    x = torch.randn(4, C, 10, 10, device=DEV)
    x1 = x.clone().requires_grad_(True)
    x2 = x.clone().to(memory_format=torch.channels_last).requires_grad_(True)
    y1 = torch.relu(ref(x1))
    y2 = fused(x2)
    assert torch.allclose(y1, y2, rtol=1e-4, atol=1e-5)
    g = torch.randn_like(y1)
    y1.backward(g)
    y2.backward(g.to(memory_format=torch.channels_last))
    assert torch.allclose(x1.grad, x2.grad, rtol=1e-4, atol=1e-5), \
        (x1.grad - x2.grad).abs().max()
    assert torch.allclose(ref.weight.grad, fused.weight.grad,
                          rtol=1e-4, atol=1e-4)


def test_process_set_gpu(hvd):
    """Per-set RCCL communicator creation (ws=1: set {0} gets its own comm
    distinct from the global one)."""
    ps = hvd.add_process_set([0])
    t = torch.arange(1000, dtype=torch.float32, device=DEV)
    out = hvd.allreduce(t, op=hvd.Sum, process_set=ps, name="psg")
    assert torch.allclose(out, t)
    out2 = hvd.allreduce(t * 2, op=hvd.Sum, name="psg_global")
    assert torch.allclose(out2, t * 2)


def test_fused_sgd_channels_last_gpu(hvd):
    """channels_last params/grads (ResNet NHWC path): the flat-storage
    kernels must accept matching non-contiguous dense layouts."""
    import torch.nn as nn

    from axon.optim import FusedSGD

    torch.manual_seed(11)
    m1 = nn.Sequential(nn.Conv2d(8, 16, 3, padding=1), nn.ReLU(),
                       nn.Conv2d(16, 8, 3, padding=1)).to(DEV)
    m2 = nn.Sequential(nn.Conv2d(8, 16, 3, padding=1), nn.ReLU(),
                       nn.Conv2d(16, 8, 3, padding=1)).to(DEV)
    m2.load_state_dict(m1.state_dict())
    m1 = m1.to(memory_format=torch.channels_last)
    o1 = FusedSGD(m1.parameters(), lr=0.05, momentum=0.9, weight_decay=1e-3)
    o2 = torch.optim.SGD(m2.parameters(), lr=0.05, momentum=0.9,
                         weight_decay=1e-3)
    for s in range(4):
        torch.manual_seed(20 + s)
        x = torch.randn(4, 8, 16, 16, device=DEV)
        (m1(x.to(memory_format=torch.channels_last)) ** 2).mean().backward()
        (m2(x) ** 2).mean().backward()
        o1.step()
        o2.step()
        o1.zero_grad()
        o2.zero_grad()
    for p, q in zip(m1.parameters(), m2.parameters()):
        if p.dim() == 4:  # conv weights stayed channels_last through step()
            assert p.is_contiguous(memory_format=torch.channels_last)
        assert torch.allclose(p.contiguous(), q.contiguous(),
                              rtol=1e-4, atol=1e-5)


# ----------------------------------------------------------- end to end

def test_resnet50_bf16_step(hvd):
    import torch.nn.functional as F

    from axon.models.resnet import resnet50
    from axon.optim import FusedSGD

    torch.manual_seed(0)
    model = resnet50().to(DEV, torch.bfloat16)
    opt = FusedSGD(model.parameters(), lr=0.1, momentum=0.9)
    opt = hvd.DistributedOptimizer(opt, named_parameters=model.named_parameters())
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)
    x = torch.randn(8, 3, 224, 224, device=DEV, dtype=torch.bfloat16)
    y = torch.randint(0, 1000, (8,), device=DEV)
    losses = []
    for _ in range(3):
        opt.zero_grad()
        loss = F.cross_entropy(model(x).float(), y)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert all(torch.isfinite(torch.tensor(losses)))
    assert losses[-1] < losses[0]  # overfits one batch fast


def test_fused_batchnorm_add_relu(hvd):
    """bn3-tail fusion: relu(bn(x) + residual) in one pass, incl. dresidual."""
    from axon.ops import FusedBatchNorm2d

    torch.manual_seed(5)
    C = 64
    ref = torch.nn.BatchNorm2d(C).to(DEV)
    fused = FusedBatchNorm2d(C).to(DEV)
    fused.load_state_dict(ref.state_dict())
    x = torch.randn(4, C, 8, 8, device=DEV)
    res = torch.randn(4, C, 8, 8, device=DEV)
    x1 = x.clone().requires_grad_(True)
    r1 = res.clone().requires_grad_(True)
    x2 = x.clone().to(memory_format=torch.channels_last).requires_grad_(True)
    r2 = res.clone().to(memory_format=torch.channels_last).requires_grad_(True)
    y1 = torch.relu(ref(x1) + r1)
    y2 = fused.forward_add_relu(x2, r2)
    assert torch.allclose(y1, y2, rtol=1e-4, atol=1e-5)
    g = torch.randn_like(y1)
    y1.backward(g)
    y2.backward(g.to(memory_format=torch.channels_last))
    assert torch.allclose(x1.grad, x2.grad, rtol=1e-4, atol=1e-5)
    assert torch.allclose(r1.grad, r2.grad, rtol=1e-4, atol=1e-5)
    assert torch.allclose(ref.weight.grad, fused.weight.grad,
                          rtol=1e-4, atol=1e-4)


@pytest.mark.parametrize("dtype,tol", [(torch.float32, 1e-5),
                                       (torch.bfloat16, 2e-2)])
@pytest.mark.parametrize("H", [64, 2048, 4096])
def test_fused_rmsnorm_matches_eager(hvd, dtype, tol, H):
    from axon.ops.rmsnorm import FusedRMSNorm

    torch.manual_seed(H)
    m = FusedRMSNorm(H).to(DEV, dtype)
    with torch.no_grad():
        m.weight.mul_(torch.randn_like(m.weight) * 0.1 + 1.0)
    rows = 37
    x = torch.randn(rows, H, device=DEV, dtype=dtype) * 2
    x1 = x.clone().requires_grad_(True)
    x2 = x.clone().requires_grad_(True)
    # eager fp32 reference (same formula the CPU fallback uses)
    xf = x1.float()
    y1 = (xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + m.eps) *
          m.weight.float()).to(dtype)
    y2 = m(x2)
    assert torch.allclose(y1, y2, rtol=tol, atol=tol), (y1 - y2).abs().max()
    g = torch.randn_like(y1)
    y1.backward(g)
    w1 = m.weight.grad.clone()
    m.weight.grad = None
    y2.backward(g)
    assert torch.allclose(x1.grad, x2.grad, rtol=tol, atol=tol * 4), \
        (x1.grad - x2.grad).abs().max()
    assert torch.allclose(w1, m.weight.grad, rtol=tol * 4, atol=tol * 40), \
        (w1 - m.weight.grad).abs().max()


def test_llama_tiny_step_gpu(hvd):
    """Tiny Llama end-to-end step on GPU through the fused RMSNorm path."""
    from axon.models.llama import LlamaForCausalLM, llama_tiny_config
    from axon.optim import FusedAdamW

    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_tiny_config()).to(DEV, torch.bfloat16)
    opt = FusedAdamW(model.parameters(), lr=1e-3)
    ids = torch.randint(0, 256, (2, 32), device=DEV)
    losses = []
    for _ in range(3):
        opt.zero_grad()
        loss = model.lm_loss(ids)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert all(torch.isfinite(torch.tensor(losses)))
    assert losses[-1] < losses[0]


def test_fused_rope_matches_eager(hvd):
    from axon.ops.rotary import _rope_eager, rope

    torch.manual_seed(9)
    b, s, h, d = 2, 33, 4, 64
    cos = torch.randn(s, d // 2, device=DEV).clamp(-1, 1)
    sin = torch.randn(s, d // 2, device=DEV).clamp(-1, 1)
    x = torch.randn(b, s, h, d, device=DEV)
    x1 = x.clone().requires_grad_(True)
    x2 = x.clone().requires_grad_(True)
    y1 = _rope_eager(x1, cos, sin)
    y2 = rope(x2, cos, sin)
    assert torch.allclose(y1, y2, rtol=1e-5, atol=1e-6), (y1 - y2).abs().max()
    g = torch.randn_like(y1)
    y1.backward(g)
    y2.backward(g)
    assert torch.allclose(x1.grad, x2.grad, rtol=1e-5, atol=1e-6)


def test_fused_swiglu_matches_eager(hvd):
    import torch.nn.functional as F

    from axon.ops.rotary import swiglu

    torch.manual_seed(10)
    a = torch.randn(1000, 48, device=DEV) * 2
    g = torch.randn(1000, 48, device=DEV)
    a1, g1 = a.clone().requires_grad_(True), g.clone().requires_grad_(True)
    a2, g2 = a.clone().requires_grad_(True), g.clone().requires_grad_(True)
    z1 = F.silu(a1) * g1
    z2 = swiglu(a2, g2)
    assert torch.allclose(z1, z2, rtol=1e-5, atol=1e-6)
    gr = torch.randn_like(z1)
    z1.backward(gr)
    z2.backward(gr)
    assert torch.allclose(a1.grad, a2.grad, rtol=1e-5, atol=1e-5)
    assert torch.allclose(g1.grad, g2.grad, rtol=1e-5, atol=1e-6)


def test_syncbn_gpu_ws1(hvd):
    """SyncBatchNorm on GPU (ws=1: must equal plain BN)."""
    from axon.sync_batch_norm import SyncBatchNorm

    torch.manual_seed(21)
    ref = torch.nn.BatchNorm2d(16).to(DEV)
    sbn = SyncBatchNorm(16).to(DEV)
    sbn.load_state_dict(ref.state_dict())
    x = torch.randn(4, 16, 6, 6, device=DEV)
    x1 = x.clone().requires_grad_(True)
    x2 = x.clone().requires_grad_(True)
    y1, y2 = ref(x1), sbn(x2)
    assert torch.allclose(y1, y2, rtol=1e-4, atol=1e-5)
    g = torch.randn_like(y1)
    y1.backward(g)
    y2.backward(g)
    assert torch.allclose(x1.grad, x2.grad, rtol=1e-4, atol=1e-5)
    assert torch.allclose(ref.running_mean, sbn.running_mean,
                          rtol=1e-4, atol=1e-5)


def test_bench_adasum_step_gpu(hvd):
    """Config-5 sanity: a ResNet step trains under op=Adasum on GPU."""
    import torch.nn.functional as F

    from axon.models.resnet import resnet50
    from axon.optim import FusedSGD

    torch.manual_seed(2)
    model = resnet50().to(DEV, torch.bfloat16)
    opt = FusedSGD(model.parameters(), lr=0.05, momentum=0.9)
    opt = hvd.DistributedOptimizer(opt, named_parameters=model.named_parameters(),
                                   op=hvd.Adasum)
    x = torch.randn(4, 3, 64, 64, device=DEV, dtype=torch.bfloat16)
    y = torch.randint(0, 1000, (4,), device=DEV)
    for _ in range(2):
        opt.zero_grad()
        loss = F.cross_entropy(model(x).float(), y)
        loss.backward()
        opt.step()
    assert torch.isfinite(loss.float())


@pytest.mark.parametrize("dtype,tol", [(torch.float32, 1e-5),
                                       (torch.bfloat16, 2e-2)])
@pytest.mark.parametrize("H", [256, 1024])
def test_fused_layernorm_matches_torch(hvd, dtype, tol, H):
    from axon.ops import FusedLayerNorm

    torch.manual_seed(H)
    ref = torch.nn.LayerNorm(H).to(DEV, dtype)
    fused = FusedLayerNorm(H).to(DEV, dtype)
    fused.load_state_dict(ref.state_dict())
    with torch.no_grad():
        for m in (ref, fused):
            m.weight.copy_(torch.randn(H).to(dtype) * 0.1 + 1)
            m.bias.copy_(torch.randn(H).to(dtype) * 0.1)
    fused.load_state_dict(ref.state_dict())
    x = torch.randn(57, H, device=DEV, dtype=dtype) * 2 + 1
    x1 = x.clone().requires_grad_(True)
    x2 = x.clone().requires_grad_(True)
    y1, y2 = ref(x1), fused(x2)
    assert torch.allclose(y1, y2, rtol=tol, atol=tol), (y1 - y2).abs().max()
    g = torch.randn_like(y1)
    y1.backward(g)
    y2.backward(g)
    assert torch.allclose(x1.grad, x2.grad, rtol=tol, atol=tol * 4), \
        (x1.grad - x2.grad).abs().max()
    assert torch.allclose(ref.weight.grad, fused.weight.grad,
                          rtol=tol * 4, atol=tol * 40)
    assert torch.allclose(ref.bias.grad, fused.bias.grad,
                          rtol=tol * 4, atol=tol * 40)


def test_bert_tiny_step_gpu(hvd):
    from axon.models.bert import BertForPreTraining, bert_large_config
    from axon.optim import FusedLAMB

    cfg = bert_large_config()
    cfg.layers, cfg.hidden, cfg.heads, cfg.ffn = 2, 128, 4, 256
    torch.manual_seed(0)
    model = BertForPreTraining(cfg).to(DEV, torch.bfloat16)
    opt = FusedLAMB(model.parameters(), lr=1e-3)
    ids = torch.randint(0, 1000, (4, 32), device=DEV)
    labels = ids.clone()
    losses = []
    for _ in range(3):
        opt.zero_grad()
        loss = model.mlm_loss(ids, labels)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert all(torch.isfinite(torch.tensor(losses)))
    assert losses[-1] < losses[0]


def test_fa_fwd_experimental_matches_sdpa(hvd):
    """Experimental CDNA4 flash-attention forward (docs/FA_PLAN.md v0,
    probe-only — not wired into models): matches SDPA on causal GQA,
    including the spiked-K rescale case and a tile-ragged length."""
    from axon import _core

    for kind, s in (("random", 192), ("random", 512), ("spiked", 256)):
        torch.manual_seed(s)
        q = torch.randn(2, 8, s, 128).clamp(-2, 2)
        k = torch.randn(2, 2, s, 128).clamp(-2, 2)
        v = torch.randn(2, 2, s, 128).clamp(-2, 2)
        if kind == "spiked":
            k[:, :, s // 2, :] = 6.0
            q[:, :, 3 * s // 4, :] = 6.0
        q, k, v = (t.to(DEV, torch.bfloat16) for t in (q, k, v))
        with torch.no_grad():
            out, lse = _core.fa_fwd(q, k, v, True)
            ref = torch.nn.functional.scaled_dot_product_attention(
                q, k, v, is_causal=True, enable_gqa=True)
        assert torch.allclose(out.float(), ref.float(), rtol=3e-2, atol=3e-2), \
            (kind, s, (out.float() - ref.float()).abs().max())
        assert torch.isfinite(lse).all()
