"""Config 1 (BASELINE.json): MNIST-shaped LeNet + SGD on CPU, ws=2.
Run: python -m axon.runner.run -np 2 --no-pin python examples/train_mnist.py
(Synthetic MNIST-shaped data: this environment has no dataset downloads;
swap _synthetic_loader for torchvision.datasets.MNIST when available.)
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

import axon as hvd
from axon.models.lenet import LeNet


def _synthetic_loader(rank, size, batches=30, bs=32, seed=0):
    for s in range(batches):
        g = torch.Generator().manual_seed(seed + s)
        y = torch.randint(0, 10, (bs,), generator=g)
        x = 0.3 * torch.randn(bs, 1, 28, 28, generator=g) \
            + (y.float().view(-1, 1, 1, 1) - 4.5) / 3.0
        yield x[rank::size], y[rank::size]  # DistributedSampler-style shard


def main():
    hvd.init()
    torch.manual_seed(42)
    model = LeNet()
    opt = torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9)
    opt = hvd.DistributedOptimizer(opt, named_parameters=model.named_parameters())
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)

    for step, (x, y) in enumerate(_synthetic_loader(hvd.rank(), hvd.size())):
        opt.zero_grad()
        loss = F.cross_entropy(model(x), y)
        loss.backward()
        opt.step()
        if hvd.rank() == 0 and step % 10 == 0:
            print(f"step {step} loss {loss.item():.4f}", flush=True)

    if hvd.rank() == 0:
        torch.save({"model": model.state_dict(),
                    "optimizer": opt.state_dict()}, "/tmp/mnist_ckpt.pt")
        print("saved checkpoint")
    hvd.shutdown()


if __name__ == "__main__":
    main()
