"""Worker for the discovery-driven elastic resize tests: commits every
step (the membership-change boundary) and records its final world view."""
import json
import os
import sys
import time

import torch

import axon as hvd
from axon import elastic

OUT = sys.argv[1]
TOTAL = int(sys.argv[2]) if len(sys.argv) > 2 else 60

torch.manual_seed(0)
model = torch.nn.Linear(4, 2)
opt = torch.optim.SGD(model.parameters(), lr=0.01)
state = elastic.TorchState(model=model, optimizer=opt, step=0, sizes=[])


@elastic.run
def train(state):
    if hvd.size() not in state.sizes:
        state.sizes = state.sizes + [hvd.size()]
    while state.step < TOTAL:
        if state.step == 5 and hvd.rank() == 0:
            # deterministic hook for the test: training is demonstrably
            # under way in THIS world size
            with open(f"{OUT}.started.{hvd.size()}", "w") as f:
                f.write("x")
        torch.manual_seed(1000 + state.step)
        x = torch.randn(6, 4)
        opt.zero_grad()
        (model(x) ** 2).mean().backward()
        for i, p in enumerate(model.parameters()):
            p.grad.copy_(hvd.allreduce(p.grad, average=True, name=f"p{i}"))
        opt.step()
        state.step += 1
        state.commit()
        time.sleep(0.05)


train(state)
wid = os.environ.get("AXON_WORKER_ID", "0")
with open(f"{OUT}.w{wid}.json", "w") as f:
    json.dump({"rank": hvd.rank(), "size": hvd.size(), "step": state.step,
               "sizes": state.sizes,
               "wsum": float(sum(p.double().sum()
                                 for p in model.parameters()))}, f)
