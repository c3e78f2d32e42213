"""Process sets: collectives over rank subgroups (SURVEY.md §2.1
"Process sets"). Covers the member path AND the rank-0 relay path (sets
that exclude rank 0, whose CPU data still star-routes through rank 0)."""
import torch

from conftest import run_distributed


def _subset_allreduce(rank, size):
    import axon as hvd

    evens = hvd.add_process_set([r for r in range(size) if r % 2 == 0])
    odds = hvd.add_process_set([r for r in range(size) if r % 2 == 1])
    mine = evens if rank % 2 == 0 else odds
    other = odds if rank % 2 == 0 else evens
    assert mine.included() and not other.included()
    assert mine.rank() == mine.ranks.index(rank)

    t = torch.full((9,), float(rank + 1))
    out = hvd.allreduce(t, op=hvd.Sum, process_set=mine, name="ps_sum")
    expect = sum(r + 1 for r in mine.ranks)
    assert torch.allclose(out, torch.full((9,), float(expect))), out

    # average divides by the SET size, not the world size
    out = hvd.allreduce(t, average=True, process_set=mine, name="ps_avg")
    assert torch.allclose(out, torch.full((9,), expect / mine.size()))

    # same user op name on both sets must not collide
    out = hvd.allreduce(t, op=hvd.Max, process_set=mine, name="shared")
    assert torch.allclose(out, torch.full((9,), float(max(mine.ranks) + 1)))


def test_process_set_allreduce_ws3():
    # odds = {1} (trivial), evens = {0, 2}
    run_distributed(_subset_allreduce, 3)


def test_process_set_allreduce_ws4():
    # odds = {1, 3}: rank 0 acts as pure relay
    run_distributed(_subset_allreduce, 4)


def _subset_no_rank0(rank, size):
    """Sets excluding rank 0 exercise the hub-relay path for every op."""
    import axon as hvd

    ps = hvd.add_process_set(list(range(1, size)))
    if rank == 0:
        # Rank 0 is only the relay here; it meets the others at the global
        # barrier at the end (global ops need every rank).
        hvd.barrier(name="nr0.done")
        return
    k = ps.rank()  # 0-based set index

    # allreduce
    out = hvd.allreduce(torch.full((5,), float(rank)), op=hvd.Sum,
                        process_set=ps, name="nr0.sum")
    assert torch.allclose(out, torch.full((5,), float(sum(ps.ranks))))

    # broadcast from the set's first member (global rank 1)
    t = torch.full((6,), float(rank))
    out = hvd.broadcast(t, root_rank=1, process_set=ps, name="nr0.bc")
    assert torch.allclose(out, torch.full((6,), 1.0))

    # variable-first-dim allgather
    t = torch.full((k + 1, 2), float(rank))
    out = hvd.allgather(t, process_set=ps, name="nr0.ag")
    rows = sum(i + 1 for i in range(ps.size()))
    assert out.shape == (rows, 2)
    off = 0
    for i, r in enumerate(ps.ranks):
        assert torch.allclose(out[off:off + i + 1], torch.full((i + 1, 2), float(r)))
        off += i + 1

    # alltoall (even splits)
    n = ps.size()
    t = torch.arange(n * 3, dtype=torch.float32).reshape(n, 3) + 100 * rank
    out = hvd.alltoall(t, process_set=ps, name="nr0.a2a")
    assert out.shape == (n, 3)
    for src in range(n):
        expect = torch.arange(3, dtype=torch.float32) + 3 * k + \
            100 * ps.ranks[src]
        assert torch.allclose(out[src], expect)

    # reducescatter (uneven rows OK)
    rows = 2 * n + 1
    t = torch.full((rows, 2), float(rank))
    out = hvd.reducescatter(t, op=hvd.Sum, process_set=ps, name="nr0.rs")
    my_rows = rows // n + (1 if k < rows % n else 0)
    assert out.shape == (my_rows, 2)
    assert torch.allclose(out, torch.full((my_rows, 2),
                                          float(sum(ps.ranks))))

    # the world is still usable for global ops afterwards
    hvd.barrier(name="nr0.done")


def test_process_set_without_rank0_ws3():
    run_distributed(_subset_no_rank0, 3)


def _subset_distopt(rank, size):
    """DistributedOptimizer over a process set: members train in lockstep,
    non-members train independently."""
    import axon as hvd

    ps = hvd.add_process_set([0, 1])
    torch.manual_seed(7)
    model = torch.nn.Linear(4, 3)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    opt = hvd.DistributedOptimizer(opt,
                                   named_parameters=model.named_parameters(),
                                   process_set=ps)
    if ps.included():
        hvd.broadcast_parameters(model.state_dict(), root_rank=0,
                                 process_set=ps)
    torch.manual_seed(100 + (rank if not ps.included() else 0))
    for s in range(3):
        torch.manual_seed(s * 10 + (0 if ps.included() else rank))
        x = torch.randn(8, 4) * (1.0 if ps.included() else rank + 1)
        opt.zero_grad()
        (model(x) ** 2).mean().backward()
        opt.step()
    # set members end bitwise identical; verified via set allgather
    if ps.included():
        w = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
        allw = hvd.allgather(w.unsqueeze(0), process_set=ps, name="wchk")
        assert torch.equal(allw[0], allw[1])
    hvd.barrier(name="distopt.done")


def test_process_set_distributed_optimizer_ws3():
    run_distributed(_subset_distopt, 3)


def _subset_errors(rank, size):
    import axon as hvd

    ps = hvd.ProcessSet([0, 1])
    try:
        hvd.allreduce(torch.ones(3), process_set=ps)
        raise AssertionError("unregistered set must raise")
    except ValueError:
        pass
    ps = hvd.add_process_set(ps)
    assert hvd.process_set_ranks(ps.process_set_id) == [0, 1]
    hvd.remove_process_set(ps)
    assert ps.process_set_id is None
    hvd.barrier(name="err.done")


def test_process_set_errors():
    run_distributed(_subset_errors, 2)


def _removed_set_op_errors(rank, size):
    import axon as hvd

    ps = hvd.add_process_set(hvd.ProcessSet(list(range(size))))
    pid = ps.process_set_id
    hvd.remove_process_set(ps)
    # Ops submitted with the raw (now removed) id must error promptly, not
    # deadlock until the stall warning (regression: ADVICE.md r1 finding).
    try:
        hvd.allreduce(torch.ones(3), name="rm.ar", process_set=pid)
        raise AssertionError("op on removed set must raise")
    except AssertionError:
        raise
    except Exception as e:  # noqa: BLE001 - surface is a core RuntimeError
        assert "removed" in str(e), str(e)
    hvd.barrier(name="rm.done")


def test_removed_set_op_errors():
    run_distributed(_removed_set_op_errors, 2)
