"""DistributedOptimizer semantics on the CPU path: hooks
fire in-place allreduces during backward; step() synchronizes; result must
equal a single-process run on the concatenated batch.
"""
import pytest
import torch
import torch.nn as nn

from conftest import run_distributed


def _make_model(seed=123):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(6, 8), nn.ReLU(), nn.Linear(8, 3))


def _data_for(rank, bs=4, seed=999):
    g = torch.Generator().manual_seed(seed + rank)
    return torch.randn(bs, 6, generator=g), torch.randn(bs, 3, generator=g)


def _reference_step(lr=0.1, ranks=(0, 1), steps=3):
    """Single-process: full batch = concat of both ranks' shards."""
    model = _make_model()
    opt = torch.optim.SGD(model.parameters(), lr=lr, momentum=0.9)
    for _ in range(steps):
        xs, ys = zip(*[_data_for(r) for r in ranks])
        x, y = torch.cat(xs), torch.cat(ys)
        opt.zero_grad()
        loss = nn.functional.mse_loss(model(x), y)
        loss.backward()
        opt.step()
    return [p.detach().clone() for p in model.parameters()]


def _dist_sgd(rank, size, lr=0.1, steps=3):
    import axon as hvd

    model = _make_model()
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)
    opt = torch.optim.SGD(model.parameters(), lr=lr, momentum=0.9)
    opt = hvd.DistributedOptimizer(
        opt, named_parameters=model.named_parameters())
    assert isinstance(opt, torch.optim.SGD)  # re-classing keeps isinstance
    for _ in range(steps):
        x, y = _data_for(rank)
        opt.zero_grad()
        loss = nn.functional.mse_loss(model(x), y)
        loss.backward()
        opt.step()
    ref = _reference_step(lr=lr, steps=steps)
    for p, r in zip(model.parameters(), ref):
        assert torch.allclose(p, r, rtol=1e-5, atol=1e-6), (p - r).abs().max()


def test_distributed_sgd_matches_single_process():
    run_distributed(_dist_sgd, 2)


def _dist_accumulate(rank, size):
    import axon as hvd

    model = _make_model()
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)
    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    opt = hvd.DistributedOptimizer(opt, backward_passes_per_step=2)
    # two backwards per step: grads accumulate locally, reduce on the 2nd.
    x1, y1 = _data_for(rank, seed=5)
    x2, y2 = _data_for(rank, seed=17)
    opt.zero_grad()
    nn.functional.mse_loss(model(x1), y1).backward()
    assert not opt._handles  # no allreduce yet
    nn.functional.mse_loss(model(x2), y2).backward()
    assert opt._handles
    opt.step()

    # reference: average of summed grads across both ranks
    ref = _make_model()
    grads = {}
    for r in range(size):
        m = _make_model()
        a1, b1 = _data_for(r, seed=5)
        a2, b2 = _data_for(r, seed=17)
        loss = nn.functional.mse_loss(m(a1), b1) + nn.functional.mse_loss(m(a2), b2)
        loss.backward()
        for n, p in m.named_parameters():
            grads[n] = grads.get(n, 0) + p.grad / size
    with torch.no_grad():
        for n, p in ref.named_parameters():
            p -= 0.05 * grads[n]
    for p, r in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(p, r, rtol=1e-5, atol=1e-6)


def test_backward_passes_per_step():
    run_distributed(_dist_accumulate, 2)


def _dist_predivide(rank, size):
    import axon as hvd

    model = _make_model()
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)
    opt = hvd.DistributedOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.1),
        gradient_predivide_factor=2.0)
    x, y = _data_for(rank)
    opt.zero_grad()
    nn.functional.mse_loss(model(x), y).backward()
    opt.step()  # prescale 1/2, sum, postscale 2/size == average
    ref = _reference_step(lr=0.1, steps=1)
    # _reference_step uses momentum; rebuild without
    ref_model = _make_model()
    g = {}
    for r in range(size):
        m = _make_model()
        a, b = _data_for(r)
        nn.functional.mse_loss(m(a), b).backward()
        for n, p in m.named_parameters():
            g[n] = g.get(n, 0) + p.grad / size
    with torch.no_grad():
        for n, p in ref_model.named_parameters():
            p -= 0.1 * g[n]
    for p, r in zip(model.parameters(), ref_model.parameters()):
        assert torch.allclose(p, r, rtol=1e-5, atol=1e-6)


def test_gradient_predivide():
    run_distributed(_dist_predivide, 2)


def _dist_compressed(rank, size):
    import axon as hvd

    model = _make_model()
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)
    opt = hvd.DistributedOptimizer(
        torch.optim.SGD(model.parameters(), lr=0.1),
        compression=hvd.Compression.fp16)
    for _ in range(2):
        x, y = _data_for(rank)
        opt.zero_grad()
        nn.functional.mse_loss(model(x), y).backward()
        opt.step()
    # fp16 wire: close to the exact run, not bitwise
    ref = _reference_step(lr=0.1, steps=2)
    ref_model = _make_model()
    # reference without momentum
    for step in range(2):
        g = {}
        for r in range(size):
            m = _make_model()
            m.load_state_dict(ref_model.state_dict())
            a, b = _data_for(r)
            nn.functional.mse_loss(m(a), b).backward()
            for n, p in m.named_parameters():
                g[n] = g.get(n, 0) + p.grad / size
        with torch.no_grad():
            for n, p in ref_model.named_parameters():
                p -= 0.1 * g[n]
    for p, r in zip(model.parameters(), ref_model.parameters()):
        assert torch.allclose(p, r, rtol=2e-3, atol=2e-3)


def test_compressed_optimizer():
    run_distributed(_dist_compressed, 2)


def _sparse_embedding_step(rank, size):
    import axon as hvd

    torch.manual_seed(5)
    emb = nn.Embedding(12, 4, sparse=True)
    hvd.broadcast_parameters(emb.state_dict(), root_rank=0)
    opt = torch.optim.SGD(emb.parameters(), lr=0.5)
    opt = hvd.DistributedOptimizer(opt,
                                   named_parameters=emb.named_parameters(),
                                   sparse_as_dense=True)
    idx = torch.tensor([rank, rank + 1, 3])
    loss = emb(idx).sum()
    loss.backward()
    opt.step()
    # reference: average of both ranks' dense grads at identical weights
    ref = torch.zeros(12, 4)
    for r in range(size):
        for i in [r, r + 1, 3]:
            ref[i] += 1.0
    ref /= size
    assert emb.weight.grad.layout == torch.strided
    assert torch.allclose(emb.weight.grad, ref), "sparse grad not averaged"


def test_sparse_as_dense_ws2():
    run_distributed(_sparse_embedding_step, 2)


def _sparse_without_flag_errors(rank, size):
    import axon as hvd

    emb = nn.Embedding(8, 4, sparse=True)
    opt = torch.optim.SGD(emb.parameters(), lr=0.5)
    opt = hvd.DistributedOptimizer(opt,
                                   named_parameters=emb.named_parameters())
    try:
        emb(torch.tensor([rank])).sum().backward()
        raise AssertionError("expected sparse-grad error")
    except (ValueError, RuntimeError) as e:
        assert "sparse_as_dense" in str(e)
    hvd.barrier(name="sparse.err.done")


def test_sparse_without_flag_errors_ws2():
    run_distributed(_sparse_without_flag_errors, 2)


def _amp_scaler_pattern(rank, size):
    """The reference's documented AMP recipe: scale -> backward ->
    optimizer.synchronize() -> unscale_ -> clip -> skip_synchronize() +
    scaler.step()."""
    import axon as hvd

    torch.manual_seed(3)
    model = nn.Linear(4, 2)
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    opt = hvd.DistributedOptimizer(opt,
                                   named_parameters=model.named_parameters())
    scaler = torch.amp.GradScaler("cpu", init_scale=2.0**8)
    g = torch.Generator().manual_seed(rank)
    x = torch.randn(6, 4, generator=g)
    for _ in range(3):
        opt.zero_grad()
        loss = (model(x) ** 2).mean()
        scaler.scale(loss).backward()
        opt.synchronize()
        scaler.unscale_(opt)
        torch.nn.utils.clip_grad_norm_(model.parameters(), 1.0)
        with opt.skip_synchronize():
            scaler.step(opt)
        scaler.update()
    # ranks stay in lockstep (same weights after averaged grads)
    wsum = torch.tensor([sum(float(p.double().sum())
                             for p in model.parameters())])
    mx = hvd.allreduce(wsum, op=hvd.Max, name="amp.wmax")
    mn = hvd.allreduce(wsum, op=hvd.Min, name="amp.wmin")
    assert abs(float(mx - mn)) < 1e-12


def test_amp_grad_scaler_pattern_ws2():
    run_distributed(_amp_scaler_pattern, 2)


def _accum_fp16_combo(rank, size):
    """Interaction coverage: gradient accumulation (bpps=2) + fp16 wire
    compression together — the allreduce fires on the second backward
    with the ACCUMULATED grad, cast on the wire, and the result matches
    a single-process run on all shards at fp16-wire tolerance."""
    import axon as hvd

    model = _make_model()
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)
    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    opt = hvd.DistributedOptimizer(opt,
                                   named_parameters=model.named_parameters(),
                                   backward_passes_per_step=2,
                                   compression=hvd.Compression.fp16)
    for micro in range(2):
        x, y = _data_for(rank, seed=500 + micro)
        loss = nn.functional.mse_loss(model(x), y)
        loss.backward()
    opt.step()
    opt.zero_grad()

    # single-process oracle: mean over the 4 micro-batches' grads
    ref = _make_model()
    xs, ys = [], []
    for r in range(size):
        for micro in range(2):
            x, y = _data_for(r, seed=500 + micro)
            xs.append(x)
            ys.append(y)
    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.05)
    ref_opt.zero_grad()
    for x, y in zip(xs, ys):
        # accumulated-sum-then-averaged-over-ranks == mean of per-rank sums
        (nn.functional.mse_loss(ref(x), y) / size).backward()
    ref_opt.step()
    for p, q in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(p, q, rtol=2e-3, atol=2e-3), (p - q).abs().max()


def test_accumulation_with_fp16_wire_ws2():
    run_distributed(_accum_fp16_combo, 2)
