"""Flagship benchmark: ResNet-50, synthetic ImageNet-shaped data, bf16,
data-parallel over N MI355X GPUs (BASELINE.json metric: images/sec
ResNet-50 synthetic at 1/2/4/8 GPUs).

Contract (driver): `python bench.py --gpus N --steps K --warmup W`;
for N>1 launched via torch.distributed.run with one rank per GPU. W
untimed warmup steps, then exactly K timed steps bracketed by barrier +
torch.cuda.synchronize on both sides; elapsed = MAX over ranks; rank 0
prints ONE JSON line with the whole-job aggregate images/sec.
"""
import argparse
import json
import os
import time

import torch
import torch.nn.functional as F

import axon as hvd


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--batch", type=int, default=256, help="per-GPU batch size")
    p.add_argument("--image-size", type=int, default=224)
    p.add_argument("--model", default="resnet50")
    p.add_argument("--fused-optimizer", default=None,
                   help="'sgd' to force the fused HIP SGD; default: auto "
                        "(fused on GPU, torch SGD on CPU)")
    p.add_argument("--compression", default="none", choices=["none", "fp16"])
    return p.parse_args()


def main():
    args = parse_args()
    hvd.init()
    n = hvd.size()
    rank = hvd.rank()
    use_gpu = torch.cuda.is_available()
    device = torch.device("cuda" if use_gpu else "cpu")
    dtype = torch.bfloat16 if use_gpu else torch.float32

    from axon.models.resnet import resnet50

    torch.manual_seed(1234)
    model = resnet50().to(device=device, dtype=dtype)

    lr = 0.1
    fused = args.fused_optimizer == "sgd" or (args.fused_optimizer is None and use_gpu)
    if fused:
        try:
            from axon.optim import FusedSGD

            opt = FusedSGD(model.parameters(), lr=lr, momentum=0.9,
                           weight_decay=1e-4)
        except (ImportError, RuntimeError):
            fused = False
    if not fused:
        opt = torch.optim.SGD(model.parameters(), lr=lr, momentum=0.9,
                              weight_decay=1e-4)
    compression = hvd.Compression.fp16 if args.compression == "fp16" \
        else hvd.Compression.none
    opt = hvd.DistributedOptimizer(opt, named_parameters=model.named_parameters(),
                                   compression=compression)
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)

    # Synthetic data, fixed per-GPU batch (weak scaling).
    g = torch.Generator().manual_seed(rank)
    x = torch.randn(args.batch, 3, args.image_size, args.image_size,
                    generator=g).to(device=device, dtype=dtype)
    y = torch.randint(0, 1000, (args.batch,), generator=g).to(device)

    def step():
        opt.zero_grad(set_to_none=False)
        out = model(x)
        loss = F.cross_entropy(out.float(), y)
        loss.backward()
        opt.step()
        return loss

    for _ in range(args.warmup):
        step()

    if use_gpu:
        torch.cuda.synchronize()
    hvd.barrier(name="bench.start")
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    if use_gpu:
        torch.cuda.synchronize()
    hvd.barrier(name="bench.stop")
    elapsed = time.perf_counter() - t0
    # MAX over ranks is the honest job time.
    elapsed = hvd.allreduce(torch.tensor([elapsed], dtype=torch.float64),
                            op=hvd.Max, name="bench.elapsed").item()

    if rank == 0:
        total_images = args.batch * n * args.steps
        value = total_images / elapsed
        print(json.dumps({
            "metric": "images/sec",
            "value": round(value, 2),
            "unit": "images/sec",
            "n_gpus": n,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": args.batch * n,
                "image_size": args.image_size,
                "parallelism": f"dp{n}",
                "optimizer": "fused_sgd" if fused else "torch_sgd",
                "compression": args.compression,
            },
        }), flush=True)
    hvd.shutdown()


if __name__ == "__main__":
    main()
