"""Mixed finish+fail elastic scenario (regression for ADVICE r1: a new
generation must not include workers that already exited cleanly).

Worker 1 exits 0 at step 5 mid-job. Its peers fail on the broken channel
and are respawned — into a generation that must contain only {0, 2}:
pre-fix the generation still listed worker 1 and the survivors hung
waiting for it to re-join.
"""
import json
import os
import sys

import torch

import axon as hvd
from axon import elastic

OUT = sys.argv[1]
TOTAL = 8

torch.manual_seed(0)
model = torch.nn.Linear(4, 2)
opt = torch.optim.SGD(model.parameters(), lr=0.05)
state = elastic.TorchState(model=model, optimizer=opt, step=0)


@elastic.run
def train(state):
    wid = os.environ.get("AXON_WORKER_ID", "0")
    while state.step < TOTAL:
        if (wid == "1" and state.step == 5 and
                not os.path.exists(OUT + ".done1")):
            with open(OUT + ".done1", "w") as f:
                f.write("x")
            os._exit(0)  # clean exit mid-job
        torch.manual_seed(1000 + state.step)
        x = torch.randn(8, 4)
        opt.zero_grad()
        (model(x) ** 2).mean().backward()
        for i, p in enumerate(model.parameters()):
            p.grad.copy_(hvd.allreduce(p.grad, average=True, name=f"p{i}"))
        opt.step()
        state.step += 1
        state.commit()
    with open(f"{OUT}.w{wid}.json", "w") as f:
        json.dump({"rank": hvd.rank(), "size": hvd.size(),
                   "step": state.step}, f)


train(state)
