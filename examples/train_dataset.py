"""Epoch-based training with axon.data.ShardedSampler + checkpoint/resume.

Shows the full reference-style workflow on a torch Dataset/DataLoader:
per-rank sharding without a torch.distributed process group, rank-0
checkpointing (the plain torch.save contract), and resume via
broadcast_parameters + broadcast_optimizer_state.

Run: python -m axon.runner.run -np 2 --no-pin python examples/train_dataset.py
Re-run the same command to resume from /tmp/axon_dataset_ckpt.pt.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F
from torch.utils.data import DataLoader, Dataset

import axon as hvd
from axon.data import ShardedSampler
from axon.models.lenet import LeNet

CKPT = "/tmp/axon_dataset_ckpt.pt"


class SyntheticMNIST(Dataset):
    """Class-conditional synthetic digits (no downloads in this env)."""

    def __init__(self, n=512, seed=0):
        g = torch.Generator().manual_seed(seed)
        self.y = torch.randint(0, 10, (n,), generator=g)
        self.x = 0.3 * torch.randn(n, 1, 28, 28, generator=g) \
            + (self.y.float().view(-1, 1, 1, 1) - 4.5) / 3.0

    def __len__(self):
        return self.y.numel()

    def __getitem__(self, i):
        return self.x[i], self.y[i]


def main():
    hvd.init()
    torch.manual_seed(42)
    model = LeNet()
    opt = torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9)
    opt = hvd.DistributedOptimizer(opt,
                                   named_parameters=model.named_parameters())

    start_epoch = 0
    if os.path.exists(CKPT):
        if hvd.rank() == 0:
            ck = torch.load(CKPT, weights_only=True)
            model.load_state_dict(ck["model"])
            opt.load_state_dict(ck["optimizer"])
            start_epoch = ck["epoch"] + 1
        start_epoch = hvd.broadcast_object(start_epoch, root_rank=0)
        hvd.broadcast_optimizer_state(opt, root_rank=0)
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)

    ds = SyntheticMNIST()
    sampler = ShardedSampler(ds, shuffle=True, seed=7)
    loader = DataLoader(ds, batch_size=32, sampler=sampler)

    for epoch in range(start_epoch, start_epoch + 3):
        sampler.set_epoch(epoch)
        for x, y in loader:
            opt.zero_grad()
            loss = F.cross_entropy(model(x), y)
            loss.backward()
            opt.step()
        avg = hvd.allreduce(loss.detach(), name="epoch.loss")
        if hvd.rank() == 0:
            print(f"epoch {epoch} loss {avg.item():.4f}", flush=True)
            torch.save({"model": model.state_dict(),
                        "optimizer": opt.state_dict(),
                        "epoch": epoch}, CKPT)
    hvd.shutdown()


if __name__ == "__main__":
    main()
