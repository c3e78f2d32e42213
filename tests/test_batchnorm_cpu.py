"""FusedBatchNorm2d CPU fallback: off the NHWC fast path it must behave
exactly like nn.BatchNorm2d (incl. fuse_relu applying ReLU)."""
import torch

from axon.ops import FusedBatchNorm2d


def test_cpu_fallback_matches_batchnorm():
    torch.manual_seed(0)
    ref = torch.nn.BatchNorm2d(12)
    fused = FusedBatchNorm2d(12)
    fused.load_state_dict(ref.state_dict())
    x = torch.randn(3, 12, 5, 5)
    x1 = x.clone().requires_grad_(True)
    x2 = x.clone().requires_grad_(True)
    y1, y2 = ref(x1), fused(x2)
    assert torch.allclose(y1, y2)
    y1.sum().backward()
    y2.sum().backward()
    assert torch.allclose(x1.grad, x2.grad)
    assert torch.allclose(ref.running_mean, fused.running_mean)


def test_cpu_fallback_fuse_relu():
    torch.manual_seed(1)
    fused = FusedBatchNorm2d(8, fuse_relu=True)
    x = torch.randn(2, 8, 4, 4)
    y = fused(x)
    assert (y >= 0).all()


def test_state_dict_interchanges_with_torch_bn():
    """Checkpoint compatibility: identical keys/buffers."""
    ref = torch.nn.BatchNorm2d(16)
    fused = FusedBatchNorm2d(16, fuse_relu=True)
    assert set(ref.state_dict().keys()) == set(fused.state_dict().keys())
    ref.load_state_dict(fused.state_dict())


def test_resnet_cpu_forward():
    """ResNet-50 with FusedBatchNorm2d still steps on CPU (fallback path)."""
    from axon.models.resnet import resnet50

    torch.manual_seed(0)
    m = resnet50(num_classes=10)
    x = torch.randn(2, 3, 64, 64)
    loss = m(x).square().mean()
    loss.backward()
    assert torch.isfinite(loss)
