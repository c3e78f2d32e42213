import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import axon as hvd
from axon.ops import FusedBatchNorm2d
from axon.optim import FusedLAMB

hvd.init()
dev = "cuda:0"
# LAMB on a chunky parameter set (MFMA norm reduction)
m = torch.nn.Linear(4096, 4096).to(dev)
opt = FusedLAMB(m.parameters(), lr=1e-3, weight_decay=0.01)
x = torch.randn(512, 4096, device=dev)
for _ in range(5):
    opt.zero_grad()
    (m(x) ** 2).mean().backward()
    opt.step()
# BN at a hot ResNet shape (C=64, 112x112)
bn = FusedBatchNorm2d(64, fuse_relu=True).to(dev)
xb = torch.randn(64, 64, 112, 112, device=dev).to(memory_format=torch.channels_last).requires_grad_(True)
for _ in range(5):
    y = bn(xb)
    y.backward(torch.randn_like(y))
    xb.grad = None
torch.cuda.synchronize()
hvd.shutdown()
print("probe done")
