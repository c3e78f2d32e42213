"""ShardedSampler / ElasticSampler: partition correctness without any
process group (identity comes from explicit args here; in real use from
hvd.rank()/hvd.size())."""
import torch

from axon.data import ElasticSampler, ShardedSampler


class _DS:
    def __init__(self, n):
        self.n = n

    def __len__(self):
        return self.n


def test_sharded_partition_covers_and_balances():
    n, world = 103, 4
    shards = [list(ShardedSampler(_DS(n), num_replicas=world, rank=r,
                                  shuffle=False))
              for r in range(world)]
    lens = {len(s) for s in shards}
    assert lens == {26}  # ceil(103/4), identical on every rank
    all_idx = [i for s in shards for i in s]
    assert set(all_idx) == set(range(n))  # full coverage
    # pad = wrap-around of the head (torch DistributedSampler semantics)
    assert len(all_idx) == 104


def test_sharded_drop_last():
    n, world = 103, 4
    shards = [list(ShardedSampler(_DS(n), num_replicas=world, rank=r,
                                  shuffle=False, drop_last=True))
              for r in range(world)]
    assert {len(s) for s in shards} == {25}
    assert len({i for s in shards for i in s}) == 100


def test_sharded_shuffle_epochs_agree_across_ranks():
    n, world = 50, 2
    s0 = ShardedSampler(_DS(n), num_replicas=world, rank=0, seed=7)
    s1 = ShardedSampler(_DS(n), num_replicas=world, rank=1, seed=7)
    for epoch in range(3):
        s0.set_epoch(epoch)
        s1.set_epoch(epoch)
        a, b = list(s0), list(s1)
        assert not (set(a) & set(b))  # disjoint shards of one permutation
        assert sorted(a + b) == list(range(n))
    s0.set_epoch(0)
    e0 = list(s0)
    s0.set_epoch(1)
    assert e0 != list(s0)  # different epochs shuffle differently


def test_sharded_rejects_bad_rank():
    try:
        ShardedSampler(_DS(10), num_replicas=2, rank=5)
        raise AssertionError("expected ValueError")
    except ValueError:
        pass


def test_elastic_resharded_remainder():
    n = 96
    world_a = 4
    samplers = [ElasticSampler(_DS(n), shuffle=False,
                               num_replicas=world_a, rank=r)
                for r in range(world_a)]
    # consume half the epoch at world 4 (global batch 8 -> 6 steps = 48)
    for s in samplers:
        for _ in range(6):
            s.record_batch(8)
    # one rank dies; world shrinks to 3, epoch resumes over the remainder
    survivors = samplers[:3]
    for r, s in enumerate(survivors):
        s.set_world(num_replicas=3, rank=r)
    lens = {len(s) for s in survivors}
    assert lens == {16}  # 48 remaining / 3
    shards = [list(s) for s in survivors]
    remainder = {i for s in shards for i in s}
    assert remainder == set(range(48, 96))  # exactly the unconsumed tail
    # new epoch resets progress and the full set comes back
    for s in survivors:
        s.set_epoch(1)
    assert {len(s) for s in survivors} == {32}


def test_elastic_iterates_torch_dataloader():
    n = 20
    xs = torch.arange(n)

    class TensorDS:
        def __len__(self):
            return n

        def __getitem__(self, i):
            return xs[i]

    s = ElasticSampler(TensorDS(), shuffle=True, num_replicas=2, rank=0)
    dl = torch.utils.data.DataLoader(TensorDS(), sampler=s, batch_size=5)
    seen = [b for b in dl]
    assert sum(x.numel() for x in seen) == 10
