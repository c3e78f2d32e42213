"""Elastic training example: survives rank failures via in-memory commits.

Run:  python -m axon.runner.run -np 4 --elastic --min-np 2 --no-pin \
          python examples/train_elastic.py
Kill a worker mid-run (kill -9 <pid>) and watch the job shrink or respawn
and continue from the last commit (axon/elastic.py).
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

import axon as hvd
from axon import elastic
from axon.models.lenet import LeNet

TOTAL_STEPS = 200

torch.manual_seed(0)
model = LeNet()
opt = torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9)
state = elastic.TorchState(model=model, optimizer=opt, step=0)
# Reset callbacks fire on survivors after every membership change —
# the place to re-shard samplers (axon.data.ElasticSampler.set_world)
# or rescale the learning rate to the new world size.
state.register_reset_callbacks(
    [lambda: print(f"[train] new world: ws={hvd.size()}", flush=True)])


@elastic.run
def train(state):
    while state.step < TOTAL_STEPS:
        g = torch.Generator().manual_seed(1000 + state.step)
        y = torch.randint(0, 10, (32,), generator=g)
        x = 0.3 * torch.randn(32, 1, 28, 28, generator=g) \
            + (y.float().view(-1, 1, 1, 1) - 4.5) / 3.0
        # shard by current rank/size (membership can change between commits)
        xs, ys = x[hvd.rank()::hvd.size()], y[hvd.rank()::hvd.size()]
        opt.zero_grad()
        loss = F.cross_entropy(model(xs), ys)
        loss.backward()
        for i, p in enumerate(model.parameters()):
            p.grad.copy_(hvd.allreduce(p.grad, average=True, name=f"g{i}"))
        opt.step()
        state.step += 1
        if state.step % 10 == 0:
            state.commit()
            if hvd.rank() == 0:
                print(f"step {state.step} ws={hvd.size()} "
                      f"loss {loss.item():.4f}", flush=True)


train(state)
if hvd.rank() == 0:
    print("done")
