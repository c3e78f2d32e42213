"""axon.checkpoint: rank-0 save (sync + async) and one-call distributed
restore, interchangeable with plain torch.save dicts."""
import os

import torch
import torch.nn as nn

from conftest import run_distributed


def _ws2_body(rank, size, path):
    import axon as hvd
    from axon import checkpoint

    torch.manual_seed(4)
    model = nn.Linear(5, 3)
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)
    opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
    for _ in range(2):
        opt.zero_grad()
        (model(torch.ones(2, 5)) ** 2).mean().backward()
        for i, p in enumerate(model.parameters()):
            p.grad.copy_(hvd.allreduce(p.grad, name=f"g{i}"))
        opt.step()
    want = [p.detach().clone() for p in model.parameters()]
    mom = [opt.state[p]["momentum_buffer"].clone()
           for p in model.parameters()]

    wrote = checkpoint.save(path, model=model, optimizer=opt, epoch=7)
    assert wrote == (rank == 0)
    hvd.barrier(name="ck.saved")
    assert os.path.exists(path)

    # wreck local state, then restore collectively
    with torch.no_grad():
        for p in model.parameters():
            p.add_(float(rank + 1))
    opt2 = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
    extra = checkpoint.restore(path, model=model, optimizer=opt2)
    assert extra == {"epoch": 7}
    for p, w in zip(model.parameters(), want):
        assert torch.allclose(p, w), "weights not restored"
    for p, m in zip(model.parameters(), mom):
        assert torch.allclose(opt2.state[p]["momentum_buffer"], m), \
            "optimizer state not restored"

    # async save writes an equivalent, torch.load-able file
    h = checkpoint.save_async(path + ".a", model=model, epoch=8)
    assert h.wait()
    hvd.barrier(name="ck.async")
    doc = torch.load(path + ".a", map_location="cpu", weights_only=True)
    assert doc["epoch"] == 8
    for k, v in model.state_dict().items():
        assert torch.allclose(doc["model"][k], v)


def test_checkpoint_roundtrip_ws2(tmp_path):
    run_distributed(_ws2_body, 2, args=(str(tmp_path / "ck.pt"),))
