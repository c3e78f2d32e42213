"""Config 2 (BASELINE.json): ResNet-50, synthetic ImageNet-shaped data,
bf16, ring allreduce over 8x MI355X.
Run: python -m axon.runner.run -np 8 python examples/train_resnet50.py
Adasum variant (config 5): add --adasum.
"""
import argparse
import time

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

import axon as hvd
from axon.models.resnet import resnet50
from axon.optim import FusedSGD


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=256)
    ap.add_argument("--steps", type=int, default=100)
    ap.add_argument("--adasum", action="store_true")
    args = ap.parse_args()

    hvd.init()
    device = "cuda" if torch.cuda.is_available() else "cpu"
    dtype = torch.bfloat16 if device == "cuda" else torch.float32
    model = resnet50().to(device, dtype)
    if device == "cuda":
        model = model.to(memory_format=torch.channels_last)
    opt = FusedSGD(model.parameters(), lr=0.1 * hvd.size(), momentum=0.9,
                   weight_decay=1e-4)
    opt = hvd.DistributedOptimizer(
        opt, named_parameters=model.named_parameters(),
        op=hvd.Adasum if args.adasum else hvd.Average)
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)

    g = torch.Generator().manual_seed(hvd.rank())
    x = torch.randn(args.batch, 3, 224, 224, generator=g).to(device, dtype)
    if device == "cuda":
        x = x.to(memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (args.batch,), generator=g).to(device)

    t0 = time.perf_counter()
    for step in range(args.steps):
        opt.zero_grad()
        loss = F.cross_entropy(model(x).float(), y)
        loss.backward()
        opt.step()
        if hvd.rank() == 0 and step % 20 == 0:
            print(f"step {step} loss {loss.item():.3f}", flush=True)
    if device == "cuda":
        torch.cuda.synchronize()
    if hvd.rank() == 0:
        dt = time.perf_counter() - t0
        print(f"{args.batch * hvd.size() * args.steps / dt:.1f} images/sec")
    hvd.shutdown()


if __name__ == "__main__":
    main()
