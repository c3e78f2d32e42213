"""Double-failure elastic script (tests/test_elastic.py): workers 2 AND 3
kill themselves at the same step; the launcher must handle the multi-death
sweep in one generation change."""
import json
import os
import sys

import torch

import axon as hvd
from axon import elastic

OUT = sys.argv[1]
TOTAL = 10

torch.manual_seed(0)
model = torch.nn.Linear(4, 2)
opt = torch.optim.SGD(model.parameters(), lr=0.05)
state = elastic.TorchState(model=model, optimizer=opt, step=0)


@elastic.run
def train(state):
    wid = os.environ.get("AXON_WORKER_ID", "0")
    while state.step < TOTAL:
        if (wid in ("2", "3") and state.step == 4 and
                not os.path.exists(f"{OUT}.died{wid}")):
            with open(f"{OUT}.died{wid}", "w") as f:
                f.write("x")
            os._exit(9)
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
                   "step": state.step,
                   "wsum": float(sum(p.double().sum() for p in
                                     model.parameters()))}, f)


train(state)
