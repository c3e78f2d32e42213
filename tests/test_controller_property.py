"""Controller property tests (SURVEY.md §7 hard part #1): the identical-
collective-order invariant must hold under randomized submission orders,
interleavings, and op mixes. Every rank submits the same logical set of
ops but in a different (seeded-random) order with random delays; the
engine must complete all of them with correct results and no deadlock."""
import random

import torch

from conftest import run_distributed


def _random_order_ops(rank, size, seed):
    import time

    import axon as hvd

    ops = []
    rng = random.Random(seed)  # same op set on every rank
    for i in range(12):
        kind = rng.choice(["allreduce", "broadcast", "allgather"])
        numel = rng.randint(1, 2000)
        ops.append((i, kind, numel))

    order = list(range(len(ops)))
    # DIFFERENT order per rank: the controller must still agree globally.
    random.Random(seed * 131 + rank).shuffle(order)

    jitter = random.Random(seed * 17 + rank)
    handles = {}
    for idx in order:
        i, kind, numel = ops[idx]
        if jitter.random() < 0.4:
            time.sleep(jitter.random() * 0.004)
        t = torch.full((numel,), float(rank + 1))
        if kind == "allreduce":
            handles[i] = ("allreduce",
                          hvd.allreduce_async(t, op=hvd.Sum, name=f"p{i}"), numel)
        elif kind == "broadcast":
            handles[i] = ("broadcast",
                          hvd.broadcast_async(t, root_rank=i % size,
                                              name=f"p{i}"), numel)
        else:
            handles[i] = ("allgather",
                          hvd.allgather_async(t, name=f"p{i}"), numel)

    for i, (kind, h, numel) in sorted(handles.items()):
        out = hvd.synchronize(h)
        if kind == "allreduce":
            expect = float(sum(range(1, size + 1)))
            assert torch.allclose(out, torch.full((numel,), expect)), (i, kind)
        elif kind == "broadcast":
            assert torch.allclose(out, torch.full((numel,), float(i % size + 1))), \
                (i, kind)
        else:
            assert out.shape[0] == numel * size
            for r in range(size):
                assert torch.allclose(out[r * numel:(r + 1) * numel],
                                      torch.full((numel,), float(r + 1))), (i, r)
    hvd.barrier(name="prop.done")


def test_randomized_submission_orders():
    """5 seeds x ws=3: shuffled per-rank submission with jitter."""
    for seed in range(5):
        run_distributed(_random_order_ops, 3, args=(seed,))


def test_randomized_submission_orders_ws4():
    for seed in (7, 11):
        run_distributed(_random_order_ops, 4, args=(seed,))


def _cache_churn(rank, size, seed):
    """Response-cache property: repeated NAMED ops (cache-id steady state)
    with random shape/dtype churn must stay correct — every churn forces a
    fingerprint mismatch, a full re-negotiation, and a template refresh."""
    import axon as hvd

    rng = random.Random(seed)  # same decisions on every rank
    shapes = {}
    for step in range(30):
        name = f"c{rng.randint(0, 4)}"  # few names -> heavy reuse
        if name not in shapes or rng.random() < 0.3:
            shapes[name] = (rng.randint(1, 500),
                            rng.choice([torch.float32, torch.float64]))
        numel, dt = shapes[name]
        kind = rng.choice(["allreduce", "allgather"])
        t = torch.full((numel,), float(rank + 1), dtype=dt)
        if kind == "allreduce":
            out = hvd.allreduce(t, op=hvd.Sum, name=name)
            assert out.dtype == dt and out.numel() == numel
            assert torch.allclose(
                out, torch.full((numel,), float(sum(range(1, size + 1))),
                                dtype=dt)), (step, name)
        else:
            # allgather under the same name namespace would collide with
            # the allreduce entries; scope it
            out = hvd.allgather(t, name=f"ag.{name}")
            assert out.numel() == numel * size


def test_response_cache_churn_ws3():
    run_distributed(_cache_churn, 3, args=(11,))


def _mixed_set_ops(rank, size, seed):
    """Global and process-set collectives interleaved in DIFFERENT
    per-rank submission orders: set-aware readiness must keep the global
    response order consistent and every result correct (ws=3, set {0,2})."""
    import time

    import axon as hvd

    ps = hvd.add_process_set([0, 2])
    rng = random.Random(seed)
    ops = []
    for i in range(10):
        ops.append((i, rng.choice(["global", "subset"]),
                    rng.randint(1, 300)))
    order = list(range(len(ops)))
    random.Random(seed * 31 + rank).shuffle(order)
    jitter = random.Random(seed * 7 + rank)
    handles = {}
    for idx in order:
        i, kind, numel = ops[idx]
        if jitter.random() < 0.4:
            time.sleep(jitter.random() * 0.003)
        t = torch.full((numel,), float(rank + 1))
        if kind == "global":
            handles[i] = ("g", hvd.allreduce_async(t, op=hvd.Sum,
                                                   name=f"mx{i}"), numel)
        elif ps.included():
            handles[i] = ("s", hvd.allreduce_async(t, op=hvd.Sum,
                                                   name=f"mx{i}",
                                                   process_set=ps), numel)
    for i, (kind, h, numel) in sorted(handles.items()):
        out = hvd.synchronize(h)
        expect = 6.0 if kind == "g" else 4.0  # ranks 1+2+3 vs 1+3
        assert torch.allclose(out, torch.full((numel,), expect)), (i, kind)
    hvd.barrier(name="mx.done")


def test_mixed_set_randomized_orders_ws3():
    for seed in (3, 17):
        run_distributed(_mixed_set_ops, 3, args=(seed,))
