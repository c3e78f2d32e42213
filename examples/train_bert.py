"""Config 3 (BASELINE.json): BERT-large pretrain, LAMB + fp16 gradient
compression, 8x MI355X.
Run: python -m axon.runner.run -np 8 python examples/train_bert.py
"""
import argparse

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import axon as hvd
from axon.models.bert import BertForPreTraining, bert_large_config
from axon.optim import FusedLAMB


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--seq", type=int, default=128)
    ap.add_argument("--steps", type=int, default=50)
    args = ap.parse_args()

    hvd.init()
    device = "cuda" if torch.cuda.is_available() else "cpu"
    dtype = torch.bfloat16 if device == "cuda" else torch.float32
    model = BertForPreTraining(bert_large_config()).to(device, dtype)
    opt = FusedLAMB(model.parameters(), lr=1e-3, weight_decay=0.01)
    opt = hvd.DistributedOptimizer(opt, named_parameters=model.named_parameters(),
                                   compression=hvd.Compression.fp16)
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)

    g = torch.Generator().manual_seed(hvd.rank())
    ids = torch.randint(0, 30522, (args.batch, args.seq), generator=g).to(device)
    labels = ids.clone()
    labels[(torch.rand(args.batch, args.seq, generator=g) > 0.15).to(device)] = -100

    for step in range(args.steps):
        opt.zero_grad()
        loss = model.mlm_loss(ids, labels)
        loss.backward()
        opt.step()
        if hvd.rank() == 0 and step % 10 == 0:
            print(f"step {step} mlm loss {loss.item():.3f}", flush=True)
    hvd.shutdown()


if __name__ == "__main__":
    main()
