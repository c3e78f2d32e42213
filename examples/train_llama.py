"""Config 4 (BASELINE.json): Llama-3 8B pure data-parallel bf16 — 288 GB
HBM3E per MI355X holds params + grads + optimizer state with no sharding;
gradient tensors are GB-scale so the fused allreduce streams large buffers.
Run: python -m axon.runner.run -np 8 python examples/train_llama.py
"""
import argparse

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import axon as hvd
from axon.models.llama import (LlamaForCausalLM, llama3_1b_config,
                               llama3_8b_config)
from axon.optim import FusedAdamW


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=2)
    ap.add_argument("--seq", type=int, default=4096)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--small", action="store_true", help="1B variant")
    args = ap.parse_args()

    hvd.init()
    device = "cuda" if torch.cuda.is_available() else "cpu"
    dtype = torch.bfloat16 if device == "cuda" else torch.float32
    cfg = llama3_1b_config() if args.small else llama3_8b_config()
    with torch.device(device):  # on-device init (8B fp32 on host is slow)
        model = LlamaForCausalLM(cfg)
    model = model.to(dtype=dtype)
    opt = FusedAdamW(model.parameters(), lr=3e-4)
    opt = hvd.DistributedOptimizer(opt, named_parameters=model.named_parameters())
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)

    g = torch.Generator().manual_seed(hvd.rank())
    ids = torch.randint(0, cfg.vocab_size, (args.batch, args.seq),
                        generator=g).to(device)

    for step in range(args.steps):
        opt.zero_grad()
        loss = model.lm_loss(ids)
        loss.backward()
        opt.step()
        if hvd.rank() == 0 and step % 5 == 0:
            print(f"step {step} lm loss {loss.item():.3f}", flush=True)
    hvd.shutdown()


if __name__ == "__main__":
    main()
