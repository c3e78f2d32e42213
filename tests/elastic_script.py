"""Training script for the elastic fault-injection test (run under
`axonrun --elastic`, tests/test_elastic.py). Worker 2 kills itself once at
step 5; survivors must recover from their last commit and finish."""
import json
import os
import sys

import torch

import axon as hvd
from axon import elastic

OUT = sys.argv[1]
TOTAL = 12

torch.manual_seed(0)
model = torch.nn.Linear(4, 2)
opt = torch.optim.SGD(model.parameters(), lr=0.05)
state = elastic.TorchState(model=model, optimizer=opt, step=0)


def _on_reset():
    # Survivors record the membership change (reset-callback parity).
    wid = os.environ.get("AXON_WORKER_ID", "0")
    with open(f"{OUT}.reset.w{wid}", "a") as f:
        f.write(f"{hvd.size()}\n")


state.register_reset_callbacks([_on_reset])


@elastic.run
def train(state):
    wid = os.environ.get("AXON_WORKER_ID", "0")
    while state.step < TOTAL:
        if (wid == "2" and state.step == 5 and
                not os.path.exists(OUT + ".died")):
            with open(OUT + ".died", "w") as f:
                f.write("x")
            os._exit(7)
        torch.manual_seed(1000 + state.step)  # same data on all ranks
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
