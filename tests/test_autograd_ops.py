"""Differentiable collectives (hvd parity): backward of a sync collective
re-enters the matching collective, so gradients aggregate the derivative
of EVERY rank's loss. Each case checks a closed-form expectation."""
import torch

from conftest import run_distributed


def _allreduce_grads(rank, size):
    import axon as hvd

    x = torch.full((3,), float(rank + 1), requires_grad=True)
    out = hvd.allreduce(x, op=hvd.Average, name="ar.avg")
    assert out.requires_grad
    out.sum().backward()
    # backward: allreduce-average of ones over ranks -> ones
    assert torch.allclose(x.grad, torch.ones(3)), x.grad

    x2 = torch.full((3,), 1.0, requires_grad=True)
    hvd.allreduce(x2, op=hvd.Sum, name="ar.sum").sum().backward()
    assert torch.allclose(x2.grad, torch.full((3,), float(size))), x2.grad


def test_allreduce_autograd():
    run_distributed(_allreduce_grads, 2)


def _broadcast_grads(rank, size):
    import axon as hvd

    x = (torch.arange(4.0) + rank).requires_grad_()
    out = hvd.broadcast(x, root_rank=0, name="bc")
    out.sum().backward()
    if rank == 0:
        # every rank's d(sum)/d(out) = ones, summed onto the root
        assert torch.allclose(x.grad, torch.full((4,), float(size))), x.grad
    else:
        assert torch.allclose(x.grad, torch.zeros(4)), x.grad


def test_broadcast_autograd():
    run_distributed(_broadcast_grads, 2)


def _allgather_grads(rank, size):
    import axon as hvd

    x = torch.full((2, 3), float(rank + 1), requires_grad=True)
    out = hvd.allgather(x, name="ag")
    assert out.shape == (2 * size, 3)
    out.pow(2).sum().backward()
    # d(sum_r L_r)/dx = size * 2 * x (every rank's loss sees my shard)
    assert torch.allclose(x.grad, 2.0 * size * x.detach()), x.grad


def test_allgather_autograd():
    run_distributed(_allgather_grads, 2)


def _reducescatter_grads(rank, size):
    import axon as hvd

    x = torch.full((4, 2), float(rank + 1), requires_grad=True)
    out = hvd.reducescatter(x, op=hvd.Average, name="rs")
    assert out.shape == (2, 2)
    out.sum().backward()
    # each input element feeds exactly one rank's output slice with
    # coefficient 1/size -> grad = 1/size everywhere
    assert torch.allclose(x.grad, torch.full((4, 2), 1.0 / size)), x.grad


def test_reducescatter_autograd():
    run_distributed(_reducescatter_grads, 2)


def _alltoall_grads(rank, size):
    import axon as hvd

    # 2 rows to each peer; downstream weight (recipient rank + 1) must
    # flow back to the row's sender.
    x = torch.ones(2 * size, 3, requires_grad=True)
    out, received = hvd.alltoall(x, splits=[2] * size, name="a2a")
    (out * (rank + 1)).sum().backward()
    expect = torch.cat([torch.full((2, 3), float(p + 1))
                        for p in range(size)])
    assert torch.allclose(x.grad, expect), x.grad


def test_alltoall_autograd():
    run_distributed(_alltoall_grads, 2)


def _grouped_allreduce_grads(rank, size):
    import axon as hvd

    ts = [torch.full((2,), float(rank + 1 + i), requires_grad=True)
          for i in range(3)]
    outs = hvd.grouped_allreduce(ts, op=hvd.Average, name="gar")
    sum(o.sum() for o in outs).backward()
    for t in ts:
        assert torch.allclose(t.grad, torch.ones(2)), t.grad


def test_grouped_allreduce_autograd():
    run_distributed(_grouped_allreduce_grads, 2)


def _subset_autograd(rank, size):
    """Differentiable collectives over a process SET (ws=3, members
    {0, 2}): the backward must use set-relative ranks/sizes."""
    import axon as hvd

    ps = hvd.add_process_set([0, 2])
    if not ps.included():
        hvd.barrier(name="subset.done")
        return
    mem = ps.rank()  # 0 for rank0, 1 for rank2
    x = torch.full((2, 2), float(mem + 1), requires_grad=True)
    out = hvd.allreduce(x, op=hvd.Average, name="ps.ar", process_set=ps)
    assert torch.allclose(out.detach(), torch.full((2, 2), 1.5))
    out.sum().backward()
    assert torch.allclose(x.grad, torch.ones(2, 2)), x.grad

    y = torch.full((1 + mem, 3), float(mem), requires_grad=True)
    g = hvd.allgather(y, name="ps.ag", process_set=ps)
    assert g.shape == (3, 3)
    g.pow(2).sum().backward()
    # 2 members' losses each see my shard: grad = 2 * 2 * y
    assert torch.allclose(y.grad, 4.0 * y.detach()), y.grad
    hvd.barrier(name="subset.done")


def test_subset_autograd_ws3():
    run_distributed(_subset_autograd, 3)
