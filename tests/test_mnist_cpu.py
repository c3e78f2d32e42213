"""Config 1 (BASELINE.json:7): MNIST-shaped LeNet + SGD on CPU at
world_size=2 — the end-to-end plumbing config. Synthetic MNIST-shaped data
(no dataset downloads in this environment); correctness bar: the ws=2 run
with per-rank shards must match a single-process run on the full batch
(averaged gradients == full-batch gradient), and loss must decrease.
"""
import torch
import torch.nn as nn
import torch.nn.functional as F

from conftest import run_distributed


def _synthetic_mnist(n=64, seed=0):
    """Learnable MNIST-shaped data: class-conditional mean intensity plus
    noise, so the loss visibly decreases within a few steps."""
    g = torch.Generator().manual_seed(seed)
    y = torch.randint(0, 10, (n,), generator=g)
    x = 0.3 * torch.randn(n, 1, 28, 28, generator=g)
    x += (y.float().view(-1, 1, 1, 1) - 4.5) / 3.0
    return x, y


def _train(model, opt, shards, steps):
    losses = []
    for s in range(steps):
        x, y = shards[s % len(shards)]
        opt.zero_grad()
        loss = F.cross_entropy(model(x), y)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    return losses


def _mnist_ws2(rank, size):
    import axon as hvd

    from axon.models.lenet import LeNet

    torch.manual_seed(42)
    model = LeNet()
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)
    opt = torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9)
    opt = hvd.DistributedOptimizer(opt, named_parameters=model.named_parameters())

    # DistributedSampler-style even sharding of one global batch stream.
    steps = 6
    shards = []
    for s in range(steps):
        x, y = _synthetic_mnist(n=32, seed=s)
        shards.append((x[rank::size], y[rank::size]))
    _train(model, opt, shards, steps)

    # single-process reference on the full batches
    torch.manual_seed(42)
    ref = LeNet()
    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.05, momentum=0.9)
    full = [_synthetic_mnist(n=32, seed=s) for s in range(steps)]
    ref_losses = _train(ref, ref_opt, full, steps)

    for (n1, p), (n2, r) in zip(model.named_parameters(), ref.named_parameters()):
        assert torch.allclose(p, r, rtol=1e-4, atol=1e-5), \
            (n1, (p - r).abs().max().item())
    assert ref_losses[-1] < ref_losses[0]  # it actually learns


def test_mnist_lenet_ws2_matches_single_process():
    run_distributed(_mnist_ws2, 2, timeout=180)
