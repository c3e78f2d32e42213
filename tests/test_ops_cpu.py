"""CPU multi-process semantics of every collective op (SURVEY.md §4:
"every public op asserted against a locally computed expectation, run with
small world sizes"). Closed-form expectations: rank r contributes
deterministic tensors, so every rank can verify the reduction locally.
"""
import pytest
import torch

from conftest import run_distributed


# ------------------------------------------------------------- allreduce

def _allreduce_avg(rank, size):
    import axon as hvd

    t = torch.full((17,), float(rank + 1))
    out = hvd.allreduce(t, average=True)
    expect = sum(range(1, size + 1)) / size
    assert torch.allclose(out, torch.full((17,), expect))
    assert not torch.allclose(t, out) or size == 1  # out-of-place


def test_allreduce_average():
    run_distributed(_allreduce_avg, 2)


def test_allreduce_average_ws4():
    run_distributed(_allreduce_avg, 4)


def _allreduce_ops(rank, size):
    import axon as hvd

    base = torch.arange(8, dtype=torch.float32) + rank
    assert torch.allclose(
        hvd.allreduce(base, op=hvd.Sum),
        torch.arange(8, dtype=torch.float32) * size + sum(range(size)))
    assert torch.allclose(hvd.allreduce(base, op=hvd.Min),
                          torch.arange(8, dtype=torch.float32))
    assert torch.allclose(hvd.allreduce(base, op=hvd.Max),
                          torch.arange(8, dtype=torch.float32) + size - 1)
    prod = torch.ones(8)
    for r in range(size):
        prod *= torch.arange(8, dtype=torch.float32) + r
    assert torch.allclose(hvd.allreduce(base, op=hvd.Product), prod)


def test_allreduce_ops():
    run_distributed(_allreduce_ops, 2)


def _allreduce_dtypes(rank, size):
    import axon as hvd

    for dt, tol in [(torch.float32, 1e-6), (torch.float64, 1e-12),
                    (torch.bfloat16, 1e-2), (torch.float16, 1e-3)]:
        t = (torch.arange(33, dtype=torch.float32) * (rank + 1)).to(dt)
        out = hvd.allreduce(t, op=hvd.Sum, name=f"dt.{dt}")
        expect = torch.zeros(33, dtype=torch.float32)
        for r in range(size):
            expect += (torch.arange(33, dtype=torch.float32) * (r + 1)).to(dt).float()
        assert out.dtype == dt
        assert torch.allclose(out.float(), expect, rtol=tol, atol=tol), dt
    t = torch.arange(10, dtype=torch.int64) + rank
    out = hvd.allreduce(t, op=hvd.Sum, name="dt.int64")
    assert torch.equal(out, torch.arange(10, dtype=torch.int64) * size + sum(range(size)))


def test_allreduce_dtypes():
    run_distributed(_allreduce_dtypes, 2)


def _allreduce_inplace(rank, size):
    import axon as hvd

    t = torch.full((5,), float(rank))
    ret = hvd.allreduce_(t, average=True)
    assert ret.data_ptr() == t.data_ptr()
    assert torch.allclose(t, torch.full((5,), sum(range(size)) / size))


def test_allreduce_inplace():
    run_distributed(_allreduce_inplace, 2)


def _allreduce_scales(rank, size):
    import axon as hvd

    t = torch.full((9,), 2.0)
    out = hvd.allreduce(t, op=hvd.Sum, prescale_factor=0.5, postscale_factor=3.0)
    assert torch.allclose(out, torch.full((9,), 2.0 * 0.5 * size * 3.0))


def test_allreduce_scales():
    run_distributed(_allreduce_scales, 2)


def _allreduce_fp16_wire(rank, size):
    import axon as hvd

    t = torch.randn(257, generator=torch.Generator().manual_seed(rank))
    out = hvd.allreduce(t, op=hvd.Sum, compression=hvd.Compression.fp16)
    expect = torch.zeros(257)
    for r in range(size):
        expect += torch.randn(257, generator=torch.Generator().manual_seed(r)).half().float()
    assert out.dtype == torch.float32
    assert torch.allclose(out, expect, rtol=2e-3, atol=2e-3)


def test_allreduce_fp16_wire():
    run_distributed(_allreduce_fp16_wire, 2)


def _allreduce_many_fused(rank, size):
    import axon as hvd

    handles = []
    for i in range(60):
        t = torch.full((128 + i,), float(rank + i))
        handles.append(hvd.allreduce_async(t, op=hvd.Sum, name=f"many.{i}"))
    for i, h in enumerate(handles):
        out = hvd.synchronize(h)
        expect = float(sum(r + i for r in range(size)))
        assert torch.allclose(out, torch.full((128 + i,), expect)), i


def test_allreduce_many_fused():
    run_distributed(_allreduce_many_fused, 2)


def _allreduce_shuffled_order(rank, size):
    # Controller ordering invariant: ranks enqueue in different orders but
    # every collective still completes correctly (SURVEY.md §7 hard part 1).
    import random

    import axon as hvd

    idx = list(range(40))
    random.Random(rank * 77).shuffle(idx)
    handles = {}
    for i in idx:
        t = torch.full((64,), float(rank * 100 + i))
        handles[i] = hvd.allreduce_async(t, op=hvd.Sum, name=f"shuf.{i}")
    for i, h in handles.items():
        out = hvd.synchronize(h)
        expect = float(sum(r * 100 + i for r in range(size)))
        assert torch.allclose(out, torch.full((64,), expect)), i


def test_allreduce_shuffled_order():
    run_distributed(_allreduce_shuffled_order, 4)


def _grouped(rank, size):
    import axon as hvd

    ts = [torch.full((11 * (i + 1),), float(rank + i)) for i in range(5)]
    outs = hvd.grouped_allreduce(ts, op=hvd.Sum)
    for i, out in enumerate(outs):
        assert torch.allclose(out, torch.full((11 * (i + 1),),
                                              float(sum(r + i for r in range(size)))))


def test_grouped_allreduce():
    run_distributed(_grouped, 2)


# ------------------------------------------------------------- broadcast

def _broadcast(rank, size):
    import axon as hvd

    for root in range(size):
        t = torch.full((7,), float(rank))
        out = hvd.broadcast(t, root_rank=root, name=f"bc.{root}")
        assert torch.allclose(out, torch.full((7,), float(root)))
        t2 = torch.full((7,), float(rank))
        hvd.broadcast_(t2, root_rank=root, name=f"bci.{root}")
        assert torch.allclose(t2, torch.full((7,), float(root)))


def test_broadcast():
    run_distributed(_broadcast, 2)


# ------------------------------------------------------------- allgather

def _allgather_equal(rank, size):
    import axon as hvd

    t = torch.full((3, 4), float(rank))
    out = hvd.allgather(t)
    assert out.shape == (3 * size, 4)
    for r in range(size):
        assert torch.allclose(out[3 * r:3 * (r + 1)], torch.full((3, 4), float(r)))


def _allgather_variable(rank, size):
    import axon as hvd

    t = torch.full((rank + 1, 2), float(rank))
    out = hvd.allgather(t)
    assert out.shape == (sum(r + 1 for r in range(size)), 2)
    off = 0
    for r in range(size):
        assert torch.allclose(out[off:off + r + 1], torch.full((r + 1, 2), float(r)))
        off += r + 1


def _allgather_zero_rows(rank, size):
    import axon as hvd

    # Rank 1 contributes zero rows — legal; must not leave a stale frame
    # queued on its channel (regression: ADVICE.md r1 medium finding).
    rows = 0 if rank == 1 else rank + 1
    t = torch.full((rows, 2), float(rank))
    out = hvd.allgather(t, name="zr.ag")
    expect_rows = sum(0 if r == 1 else r + 1 for r in range(size))
    assert out.shape == (expect_rows, 2)
    off = 0
    for r in range(size):
        n = 0 if r == 1 else r + 1
        assert torch.allclose(out[off:off + n], torch.full((n, 2), float(r)))
        off += n
    # A following collective must see a clean channel.
    s = hvd.allreduce(torch.full((4,), float(rank)), op=hvd.Sum, name="zr.after")
    assert torch.allclose(s, torch.full((4,), float(sum(range(size)))))


def test_allgather():
    run_distributed(_allgather_equal, 2)
    run_distributed(_allgather_variable, 3)
    run_distributed(_allgather_zero_rows, 2)
    run_distributed(_allgather_zero_rows, 3)


# -------------------------------------------------------------- alltoall

def _alltoall(rank, size):
    import axon as hvd

    # rank r sends (d+1) rows of value r*10+d to rank d
    splits = [d + 1 for d in range(size)]
    rows = sum(splits)
    t = torch.cat([torch.full((d + 1, 3), float(rank * 10 + d)) for d in range(size)])
    assert t.shape[0] == rows
    out, received = hvd.alltoall(t, splits=torch.tensor(splits))
    # I receive (rank+1) rows from each rank, value r*10+rank
    assert out.shape == ((rank + 1) * size, 3)
    off = 0
    for r in range(size):
        assert torch.allclose(out[off:off + rank + 1],
                              torch.full((rank + 1, 3), float(r * 10 + rank)))
        off += rank + 1
    assert received.tolist() == [rank + 1] * size


def _alltoall_even(rank, size):
    import axon as hvd

    t = torch.arange(size * 2, dtype=torch.float32) + rank * 100
    out = hvd.alltoall(t)
    expect = torch.cat([torch.arange(rank * 2, rank * 2 + 2, dtype=torch.float32) + r * 100
                        for r in range(size)])
    assert torch.allclose(out, expect)


def test_alltoall():
    run_distributed(_alltoall, 3)
    run_distributed(_alltoall_even, 2)


# --------------------------------------------------------- reducescatter

def _reducescatter(rank, size):
    import axon as hvd

    for rows in (size * 3, size * 3 + 1):  # even + uneven
        t = torch.ones(rows, 2) * (rank + 1)
        out = hvd.reducescatter(t, op=hvd.Sum, name=f"rs.{rows}")
        base, rem = rows // size, rows % size
        my_rows = base + (1 if rank < rem else 0)
        assert out.shape == (my_rows, 2)
        assert torch.allclose(out, torch.full((my_rows, 2),
                                              float(sum(range(1, size + 1)))))
        out2 = hvd.reducescatter(t, op=hvd.Average, name=f"rsa.{rows}")
        assert torch.allclose(out2, torch.full((my_rows, 2),
                                               sum(range(1, size + 1)) / size))


def test_reducescatter():
    run_distributed(_reducescatter, 2)
    run_distributed(_reducescatter, 3)


# ------------------------------------------------------------------ adasum

def _adasum_pair(a, b):
    dot = torch.dot(a, b).item()
    na = torch.dot(a, a).item()
    nb = torch.dot(b, b).item()
    ca = 1.0 - dot / (2 * na) if na > 0 else 1.0
    cb = 1.0 - dot / (2 * nb) if nb > 0 else 1.0
    return ca * a + cb * b


def adasum_oracle(vectors):
    """Reference recursive-halving adasum tree in fp64 (same tree as the
    engine: non-pow2 extras fold into partner i-p2 first, then level k
    combines v[i] with v[i+2^k])."""
    v = [x.double() for x in vectors]
    n = len(v)
    p2 = 1
    while p2 * 2 <= n:
        p2 *= 2
    for i in range(n - p2):
        v[i] = _adasum_pair(v[i], v[i + p2])
    stride = 1
    while stride < p2:
        for i in range(0, p2, 2 * stride):
            v[i] = _adasum_pair(v[i], v[i + stride])
        stride *= 2
    return v[0]


def _adasum(rank, size):
    import axon as hvd

    t = torch.randn(97, generator=torch.Generator().manual_seed(rank * 3 + 1))
    out = hvd.allreduce(t, op=hvd.Adasum)
    vectors = [torch.randn(97, generator=torch.Generator().manual_seed(r * 3 + 1))
               for r in range(size)]
    expect = adasum_oracle(vectors).float()
    assert torch.allclose(out, expect, rtol=1e-5, atol=1e-5)


def test_adasum():
    run_distributed(_adasum, 2)
    run_distributed(_adasum, 4)


def test_adasum_non_pow2():
    # Non-power-of-two sets fold extras into the low pow-2 group (VERDICT r1
    # next-round item 7).
    run_distributed(_adasum, 3)
    run_distributed(_adasum, 6)


# -------------------------------------------------------------- join etc.

def _join(rank, size):
    import axon as hvd

    # rank 0 runs 3 steps, rank 1 only 2 then joins.
    steps = 3 if rank == 0 else 2
    for i in range(steps):
        t = torch.full((6,), float(rank + 1))
        out = hvd.allreduce(t, op=hvd.Sum, name=f"j.{i}")
        if i < 2:
            assert torch.allclose(out, torch.full((6,), 3.0))
        else:
            assert torch.allclose(out, torch.full((6,), 1.0))  # peer joined -> zeros
    last = hvd.join()
    assert last in range(size)


def test_join():
    run_distributed(_join, 2)


def _barrier(rank, size):
    import axon as hvd

    for _ in range(3):
        hvd.barrier()


def test_barrier():
    run_distributed(_barrier, 4)


def _duplicate_name(rank, size):
    import time

    import axon as hvd

    # rank 0 enqueues "dup" twice before rank 1 enqueues it once: the first
    # stays pending (awaiting rank 1), so the second must get the
    # duplicate-name error deterministically.
    if rank == 0:
        h1 = hvd.allreduce_async(torch.ones(4), op=hvd.Sum, name="dup")
        h2 = hvd.allreduce_async(torch.ones(4), op=hvd.Sum, name="dup")
        try:
            hvd.synchronize(h2)
            raise AssertionError("expected duplicate-name error")
        except RuntimeError as e:
            assert "already in flight" in str(e)
        hvd.barrier()  # let rank 1 enqueue its "dup" afterwards
        hvd.synchronize(h1)
    else:
        hvd.barrier()
        time.sleep(0.05)
        hvd.allreduce(torch.ones(4), op=hvd.Sum, name="dup")


def test_duplicate_name_error():
    run_distributed(_duplicate_name, 2)


def _shape_mismatch(rank, size):
    import axon as hvd

    t = torch.ones(3 + rank)
    try:
        hvd.allreduce(t, op=hvd.Sum, name="mismatch")
        raise AssertionError("expected shape-mismatch error")
    except RuntimeError as e:
        assert "mismatch" in str(e)


def test_shape_mismatch_error():
    run_distributed(_shape_mismatch, 2)


def _scalar_tensor(rank, size):
    import axon as hvd

    t = torch.tensor(float(rank + 1))
    out = hvd.allreduce(t, average=True)
    assert out.item() == pytest.approx(sum(range(1, size + 1)) / size)


def test_scalar_tensor():
    run_distributed(_scalar_tensor, 2)


def _channels_last(rank, size):
    import axon as hvd

    t = (torch.arange(2 * 4 * 3 * 3, dtype=torch.float32).reshape(2, 4, 3, 3)
         * (rank + 1)).to(memory_format=torch.channels_last)
    out = hvd.allreduce(t, op=hvd.Sum, name="cl")
    expect = torch.arange(2 * 4 * 3 * 3, dtype=torch.float32).reshape(
        2, 4, 3, 3) * sum(r + 1 for r in range(size))
    assert torch.allclose(out, expect)
    b = (torch.ones(2, 4, 3, 3) * rank).to(memory_format=torch.channels_last)
    hvd.broadcast_(b, root_rank=1, name="clb")
    assert torch.allclose(b, torch.ones(2, 4, 3, 3))


def test_channels_last_dense_tensors():
    run_distributed(_channels_last, 2)


def _steady_state_cache(rank, size):
    # Same tensor names re-submitted across many steps: after step 1 the
    # workers ship bare cache ids (SURVEY.md §2.1 response cache) — results
    # must stay correct in every iteration.
    import axon as hvd

    for it in range(6):
        handles = [
            hvd.allreduce_async(torch.full((50 + i,), float(rank + i + it)),
                                op=hvd.Sum, name=f"steady.{i}")
            for i in range(8)
        ]
        for i, h in enumerate(handles):
            out = hvd.synchronize(h)
            expect = float(sum(r + i + it for r in range(size)))
            assert torch.allclose(out, torch.full((50 + i,), expect)), (it, i)


def test_steady_state_response_cache():
    run_distributed(_steady_state_cache, 4)


def _steady_state_cache_all_ops(rank, size):
    # Round-2 extension: broadcast / allgather /
    # reducescatter / alltoall negotiate through the response cache too.
    # Correctness must hold across repeats (bare-id path), including an
    # allgather whose per-rank first dims differ (per-rank templates) and
    # a shape change mid-stream (fingerprint invalidation).
    import axon as hvd

    for it in range(5):
        b = hvd.broadcast(torch.full((32,), float(rank * 10 + it)),
                          root_rank=1, name="sc.bc")
        assert torch.allclose(b, torch.full((32,), float(10 + it)))

        g = hvd.allgather(torch.full((rank + 1, 2), float(rank + it)),
                          name="sc.ag")
        off = 0
        for r in range(size):
            assert torch.allclose(g[off:off + r + 1],
                                  torch.full((r + 1, 2), float(r + it)))
            off += r + 1

        rows = size * 2
        rs = hvd.reducescatter(torch.full((rows, 3), float(rank + 1 + it)),
                               op=hvd.Sum, name="sc.rs")
        expect = float(sum(r + 1 + it for r in range(size)))
        assert rs.shape == (2, 3)
        assert torch.allclose(rs, torch.full((2, 3), expect))

        splits = [d + 1 for d in range(size)]
        t = torch.cat([torch.full((d + 1, 2), float(rank * 10 + d + it))
                       for d in range(size)])
        out, _ = hvd.alltoall(t, splits=torch.tensor(splits), name="sc.a2a")
        off = 0
        for r in range(size):
            assert torch.allclose(out[off:off + rank + 1],
                                  torch.full((rank + 1, 2),
                                             float(r * 10 + rank + it)))
            off += rank + 1

    # Shape change on a cached name: fingerprint mismatch must force a full
    # re-negotiation and still produce the right answer.
    g2 = hvd.allgather(torch.full((2 * rank + 1, 2), float(rank)), name="sc.ag")
    off = 0
    for r in range(size):
        assert torch.allclose(g2[off:off + 2 * r + 1],
                              torch.full((2 * r + 1, 2), float(r)))
        off += 2 * r + 1


def test_steady_state_cache_all_ops():
    run_distributed(_steady_state_cache_all_ops, 2)
    run_distributed(_steady_state_cache_all_ops, 3)


def _ws8_smoke(rank, size):
    import axon as hvd

    t = torch.full((33,), float(rank + 1))
    out = hvd.allreduce(t, op=hvd.Sum, name="ws8")
    assert torch.allclose(out, torch.full((33,), float(sum(range(1, size + 1)))))
    ps = hvd.add_process_set([0, 3, 5, 7])
    if ps.included():
        o = hvd.allreduce(t, average=True, process_set=ps, name="ws8ps")
        expect = sum(r + 1 for r in ps.ranks) / ps.size()
        assert torch.allclose(o, torch.full((33,), expect))
    hvd.barrier(name="ws8.done")


def test_world_size_8():
    """Full-width bitmask paths in the controller (size 8 = one rank per
    MI355X on the target node)."""
    run_distributed(_ws8_smoke, 8)


def _bf16_wire(rank, size):
    import axon as hvd

    t = torch.arange(64, dtype=torch.float32) * (rank + 1)
    out = hvd.allreduce(t, op=hvd.Sum, compression=hvd.Compression.bf16,
                        name="bf16wire")
    expect = torch.arange(64, dtype=torch.float32).bfloat16().float() * \
        sum(range(1, size + 1))
    assert out.dtype == torch.float32
    assert torch.allclose(out, expect, rtol=2e-2, atol=2e-2)


def test_bf16_wire_compression():
    run_distributed(_bf16_wire, 2)


def _adasum_properties(rank, size):
    """Mathematical invariants of the Adasum combine (ws=2): orthogonal
    gradients ADD (no conflict -> keep both), identical gradients AVERAGE
    (full conflict -> no double step), and the result is scale-covariant
    for a common positive factor."""
    import axon as hvd

    # orthogonal: disjoint supports
    a = torch.zeros(8, dtype=torch.float64)
    a[:4] = torch.tensor([1.0, 2.0, -1.0, 0.5], dtype=torch.float64)
    b = torch.zeros(8, dtype=torch.float64)
    b[4:] = torch.tensor([3.0, -2.0, 1.0, 1.0], dtype=torch.float64)
    mine = a if rank == 0 else b
    out = hvd.allreduce(mine.clone(), op=hvd.Adasum, name="prop.orth")
    assert torch.allclose(out, a + b, atol=1e-12), out

    # identical: average
    c = torch.tensor([1.0, -2.0, 3.0, 4.0], dtype=torch.float64)
    out2 = hvd.allreduce(c.clone(), op=hvd.Adasum, name="prop.same")
    assert torch.allclose(out2, c, atol=1e-12), out2  # avg of c and c

    # scale covariance: adasum(s*a, s*b) == s * adasum(a, b)
    g = torch.Generator().manual_seed(7 + rank)
    d = torch.randn(16, generator=g, dtype=torch.float64)
    base = hvd.allreduce(d.clone(), op=hvd.Adasum, name="prop.base")
    scaled = hvd.allreduce(3.5 * d, op=hvd.Adasum, name="prop.scaled")
    assert torch.allclose(scaled, 3.5 * base, atol=1e-10)


def test_adasum_properties_ws2():
    run_distributed(_adasum_properties, 2)


def _alltoall_zero_splits(rank, size):
    """Zero entries in the splits vector (a rank sends nothing to some
    peers, possibly nothing at all) must round-trip cleanly."""
    import axon as hvd

    # rank 0 sends 2 rows to itself only; rank 1 sends 1 row to each;
    # rank 2 sends everything to rank 0.
    table = {0: [2, 0, 0], 1: [1, 1, 1], 2: [3, 0, 0]}
    splits = table[rank]
    rows = sum(splits)
    t = torch.full((rows, 2), float(rank), dtype=torch.float32)
    out, received = hvd.alltoall(t, splits=torch.tensor(splits),
                                 name="a2a.zeros")
    expect_rows = sum(table[src][rank] for src in range(size))
    assert received.tolist() == [table[src][rank] for src in range(size)]
    assert out.shape[0] == expect_rows
    off = 0
    for src in range(size):
        n = table[src][rank]
        assert torch.all(out[off:off + n] == float(src)), (rank, src)
        off += n


def test_alltoall_zero_splits_ws3():
    run_distributed(_alltoall_zero_splits, 3)


def _degenerate_edges(rank, size):
    """Empty tensors and fewer rows than ranks must round-trip: empty
    allreduce/broadcast, reducescatter with rows < world (some ranks own
    zero rows), and an all-zero-row allgather."""
    import axon as hvd

    out = hvd.allreduce(torch.empty(0), name="e.ar")
    assert out.numel() == 0
    t = torch.full((2, 3), float(rank + 1))
    rs = hvd.reducescatter(t, op=hvd.Sum, name="e.rs")
    expect_rows = 1 if rank < 2 else 0
    assert rs.shape[0] == expect_rows
    if expect_rows:
        assert torch.allclose(rs, torch.full((1, 3), 6.0))
    g = hvd.allgather(torch.zeros(0, 4), name="e.ag")
    assert g.shape == (0, 4)
    b = hvd.broadcast(torch.empty(0), root_rank=1, name="e.bc")
    assert b.numel() == 0


def test_degenerate_edges_ws3():
    run_distributed(_degenerate_edges, 3)


def _layout_dtype_edges(rank, size):
    """Layout/dtype edges: dense-but-transposed tensors allreduce in
    place correctly (all ranks share strides -> storage order agrees);
    truly non-dense views work out-of-place via a contiguous copy and
    raise in place; integer dtypes reduce; multi-MB broadcast_object."""
    import axon as hvd

    v = (torch.arange(12.0).reshape(3, 4) + rank).t().clone().t()
    vt = v.t()
    hvd.allreduce_(vt, op=hvd.Sum, name="nc.ip")
    assert torch.allclose(vt, (torch.arange(12.0).reshape(3, 4) * 2 + 1).t())

    s = (torch.arange(8.0) + rank)[::2]
    out = hvd.allreduce(s, op=hvd.Sum, name="nd.oop")
    assert torch.allclose(out, torch.tensor([1.0, 5, 9, 13]))
    try:
        hvd.allreduce_(s, name="nd.ip")
        raise AssertionError("in-place on strided slice should raise")
    except ValueError:
        pass

    i32 = torch.tensor([1, 2, 3], dtype=torch.int32) * (rank + 1)
    assert hvd.allreduce(i32, op=hvd.Sum, name="i32").tolist() == [3, 6, 9]
    assert hvd.allreduce(torch.tensor([5 - rank]), op=hvd.Min,
                         name="i64").item() == 4

    obj = {"blob": bytes(5 << 20), "rank": 0}
    got = hvd.broadcast_object(obj if rank == 0 else None, root_rank=0)
    assert len(got["blob"]) == 5 << 20


def test_layout_dtype_edges_ws2():
    run_distributed(_layout_dtype_edges, 2)


def _alltoall_bad_splits(rank, size):
    """splits vector length != world size must surface the controller's
    validation error on every rank (not hang)."""
    import axon as hvd

    t = torch.ones(4, 2)
    try:
        hvd.alltoall(t, splits=torch.tensor([2, 2, 0]), name="bad.a2a")
        raise AssertionError("expected splits-size error")
    except RuntimeError as e:
        assert "splits" in str(e), e


def test_alltoall_bad_splits_ws2():
    run_distributed(_alltoall_bad_splits, 2)


def _broadcast_bad_root(rank, size):
    """broadcast root outside the process set errors cleanly."""
    import axon as hvd

    ps = hvd.add_process_set([0, 1])
    t = torch.ones(3)
    try:
        hvd.broadcast(t, root_rank=7, name="bad.bc", process_set=ps)
        raise AssertionError("expected bad-root error")
    except RuntimeError as e:
        assert "root" in str(e), e


def test_broadcast_bad_root_ws2():
    run_distributed(_broadcast_bad_root, 2)


def _mixed_dtype_group(rank, size):
    """Grouped allreduce over mixed dtypes: the controller fuses only
    same-(dtype, wire) runs, so results stay exact per tensor (with and
    without an fp16 wire)."""
    import axon as hvd

    ts = [torch.ones(10, dtype=torch.float32) * (rank + 1),
          torch.ones(5, dtype=torch.float64) * (rank + 1),
          torch.ones(7, dtype=torch.float32) * (rank + 1)]
    for o in hvd.grouped_allreduce(ts, op=hvd.Sum, name="mixed"):
        assert torch.allclose(o.float(), torch.full_like(o.float(), 3.0))
    for o in hvd.grouped_allreduce(ts, op=hvd.Sum, name="mixedw",
                                   compression=hvd.Compression.fp16):
        assert torch.allclose(o.float(), torch.full_like(o.float(), 3.0),
                              atol=1e-2)


def test_mixed_dtype_group_ws2():
    run_distributed(_mixed_dtype_group, 2)


def _poll_semantics(rank, size):
    """hvd.poll(handle): False while a peer withholds its submission,
    True after completion; synchronize still returns the result."""
    import time

    import axon as hvd

    t = torch.full((4,), float(rank + 1))
    if rank == 0:
        h = hvd.allreduce_async(t, op=hvd.Sum, name="pollme")
        assert hvd.poll(h) is False  # rank 1 hasn't submitted yet
        time.sleep(0.3)
        assert hvd.poll(h) is False
    hvd.barrier(name="poll.gate")
    if rank != 0:
        h = hvd.allreduce_async(t, op=hvd.Sum, name="pollme")
    deadline = time.time() + 30
    while not hvd.poll(h) and time.time() < deadline:
        time.sleep(0.01)
    assert hvd.poll(h) is True
    out = hvd.synchronize(h)
    assert torch.allclose(out, torch.full((4,), 3.0))


def test_poll_semantics_ws2():
    run_distributed(_poll_semantics, 2)
