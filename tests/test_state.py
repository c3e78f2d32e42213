"""State distribution + checkpoint/resume contract:
broadcast_parameters / broadcast_optimizer_state / broadcast_object, and a
full rank-0 torch.save checkpoint round trip.
"""
import os
import tempfile

import torch
import torch.nn as nn

from conftest import run_distributed


def _model(seed):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(4, 5), nn.Linear(5, 2))


def _bparams(rank, size):
    import axon as hvd

    model = _model(seed=100 + rank)  # divergent init
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)
    ref = _model(seed=100)  # rank 0's init
    for p, r in zip(model.parameters(), ref.parameters()):
        assert torch.equal(p, r)


def test_broadcast_parameters():
    run_distributed(_bparams, 2)


def _bobject(rank, size):
    import axon as hvd

    obj = {"epoch": 7, "lr": 0.125, "names": ["a", "b"]} if rank == 0 else None
    out = hvd.broadcast_object(obj, root_rank=0)
    assert out == {"epoch": 7, "lr": 0.125, "names": ["a", "b"]}
    outs = hvd.allgather_object({"rank": rank})
    assert outs == [{"rank": r} for r in range(size)]


def test_broadcast_object():
    run_distributed(_bobject, 2)


def _bopt_state(rank, size):
    import axon as hvd

    model = _model(seed=1)
    opt = torch.optim.Adam(model.parameters(), lr=0.01 * (rank + 1))
    # build divergent state: each rank steps on different data
    torch.manual_seed(50 + rank)
    loss = model(torch.randn(3, 4)).sum()
    loss.backward()
    opt.step()
    hvd.broadcast_optimizer_state(opt, root_rank=0)
    sd = opt.state_dict()
    ref = hvd.broadcast_object(sd if rank == 0 else None, root_rank=0,
                               name="verify")
    # states must now match rank 0 exactly
    for pid, st in ref["state"].items():
        for k, v in st.items():
            mine = sd["state"][pid][k]
            if torch.is_tensor(v):
                assert torch.equal(mine, v), (pid, k)
            else:
                assert mine == v
    for g_ref, g_mine in zip(ref["param_groups"], sd["param_groups"]):
        assert g_ref == g_mine  # includes lr from rank 0


def test_broadcast_optimizer_state():
    run_distributed(_bopt_state, 2)


def _checkpoint_roundtrip(rank, size, path):
    import axon as hvd

    model = _model(seed=3)
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)
    opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
    opt = hvd.DistributedOptimizer(opt, named_parameters=model.named_parameters())
    for step in range(2):
        opt.zero_grad()
        torch.manual_seed(rank * 10 + step)
        loss = (model(torch.randn(4, 4)) ** 2).mean()
        loss.backward()
        opt.step()

    # rank-0-only checkpoint (the contract: plain torch.save dict)
    if rank == 0:
        torch.save({"model": model.state_dict(),
                    "optimizer": opt.state_dict(), "epoch": 5}, path)
    hvd.barrier()

    # resume: fresh model, rank 0 loads, broadcast
    model2 = _model(seed=77 + rank)
    opt2 = torch.optim.SGD(model2.parameters(), lr=0.1, momentum=0.9)
    opt2 = hvd.DistributedOptimizer(opt2, named_parameters=model2.named_parameters())
    epoch = None
    if rank == 0:
        ckpt = torch.load(path, weights_only=False)
        model2.load_state_dict(ckpt["model"])
        opt2.load_state_dict(ckpt["optimizer"])
        epoch = ckpt["epoch"]
    hvd.broadcast_parameters(model2.state_dict(), root_rank=0)
    hvd.broadcast_optimizer_state(opt2, root_rank=0)
    epoch = hvd.broadcast_object(epoch, root_rank=0)
    assert epoch == 5
    for p, q in zip(model.parameters(), model2.parameters()):
        assert torch.equal(p, q)
    # momentum buffers restored everywhere
    for p, q in zip(model.parameters(), model2.parameters()):
        b1 = opt.state[p].get("momentum_buffer")
        b2 = opt2.state[q].get("momentum_buffer")
        if rank == 0:
            assert b1 is not None and torch.equal(b1, b2)
        else:
            assert b2 is not None


def test_checkpoint_roundtrip():
    with tempfile.TemporaryDirectory() as d:
        run_distributed(_checkpoint_roundtrip, 2,
                        args=(os.path.join(d, "ckpt.pt"),))


def _vanilla_interchange(rank, size, path):
    """Checkpoints interchange with stock PyTorch: a
    checkpoint written under axon loads into a plain torch optimizer."""
    import axon as hvd

    model = _model(seed=3)
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)
    opt = hvd.DistributedOptimizer(
        torch.optim.Adam(model.parameters(), lr=1e-3),
        named_parameters=model.named_parameters())
    opt.zero_grad()
    (model(torch.ones(2, 4)).sum()).backward()
    opt.step()
    if rank == 0:
        torch.save({"model": model.state_dict(), "optimizer": opt.state_dict()},
                   path)
        plain_model = _model(seed=9)
        plain_opt = torch.optim.Adam(plain_model.parameters(), lr=1e-3)
        ckpt = torch.load(path, weights_only=False)
        plain_model.load_state_dict(ckpt["model"])
        plain_opt.load_state_dict(ckpt["optimizer"])  # must not raise
        st = list(plain_opt.state.values())[0]
        assert "exp_avg" in st and "exp_avg_sq" in st and "step" in st
    hvd.barrier()


def test_vanilla_checkpoint_interchange():
    with tempfile.TemporaryDirectory() as d:
        run_distributed(_vanilla_interchange, 2,
                        args=(os.path.join(d, "ckpt.pt"),))


def _fused_opt_state_broadcast(rank, size):
    """broadcast_optimizer_state over a FUSED optimizer's state dict
    (exp_avg/exp_avg_sq/step tensors): non-root states rebuilt + equal."""
    import axon as hvd
    from axon.optim import FusedAdamW

    model = _model(seed=4)
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)
    opt = FusedAdamW(model.parameters(), lr=1e-3)
    if rank == 0:  # only root trains; others start stateless
        for _ in range(2):
            opt.zero_grad()
            (model(torch.ones(2, 4)).sum()).backward()
            opt.step()
    hvd.broadcast_optimizer_state(opt, root_rank=0)
    sums = []
    for p in model.parameters():
        st = opt.state[p]
        assert float(st["step"]) == 2
        sums.append(float(st["exp_avg"].double().sum())
                    + float(st["exp_avg_sq"].double().sum()))
    t = torch.tensor(sums, dtype=torch.float64)
    mx = hvd.allreduce(t, op=hvd.Max, name="fs.max")
    mn = hvd.allreduce(t, op=hvd.Min, name="fs.min")
    assert torch.allclose(mx, mn, atol=0, rtol=0), (mx - mn)


def test_fused_optimizer_state_broadcast():
    run_distributed(_fused_opt_state_broadcast, 2)
