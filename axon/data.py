"""Per-rank data sharding for the engine's data-parallel training loop.

Capability parity: the reference wrapper's training scripts shard their
dataset with torch's DistributedSampler (SURVEY.md §2.3 "PyTorch training
script(s) ... per-rank DistributedSampler-style sharding"). Stock
`torch.utils.data.DistributedSampler` reads rank/world from a
torch.distributed process group, which this engine never creates (RCCL is
driven by the C++ core directly), so users switching over need a sampler
that takes its identity from hvd.rank()/hvd.size() instead. Same
semantics as torch's: optional shuffle with a per-epoch seed that agrees
across ranks, padding (or dropping) so every rank sees the same number of
samples per epoch — a requirement for synchronous allreduce training,
where uneven batch counts deadlock (that's what hvd.join() is for when
padding is not acceptable).

ElasticSampler additionally re-shards on the fly when the world changes
(axon.elastic generation changes): call `set_world()` from a
@state.on_reset callback, and record progress with `record_batch()` so a
restored epoch resumes after the already-consumed samples.
"""
import math

import torch
from torch.utils.data import Sampler

from axon import basics


class ShardedSampler(Sampler):
    """DistributedSampler equivalent keyed off the engine's rank/size.

    drop_last=False pads the index list to a multiple of the world size by
    wrapping around (torch semantics), so `len()` is identical on every
    rank and collective training steps stay in lockstep.
    """

    def __init__(self, dataset, num_replicas=None, rank=None, shuffle=True,
                 seed=0, drop_last=False):
        self.dataset = dataset
        self.num_replicas = (num_replicas if num_replicas is not None
                             else basics.size())
        self.rank = rank if rank is not None else basics.rank()
        if not (0 <= self.rank < self.num_replicas):
            raise ValueError(f"rank {self.rank} outside world "
                             f"{self.num_replicas}")
        self.shuffle = shuffle
        self.seed = seed
        self.drop_last = drop_last
        self.epoch = 0
        self._recompute()

    def _recompute(self):
        n = len(self.dataset)
        if self.drop_last and n % self.num_replicas:
            self.num_samples = n // self.num_replicas
        else:
            self.num_samples = math.ceil(n / self.num_replicas)
        self.total_size = self.num_samples * self.num_replicas

    def _indices(self):
        n = len(self.dataset)
        if self.shuffle:
            g = torch.Generator()
            g.manual_seed(self.seed + self.epoch)  # same order on all ranks
            idx = torch.randperm(n, generator=g).tolist()
        else:
            idx = list(range(n))
        if self.drop_last:
            idx = idx[: self.total_size]
        else:
            while len(idx) < self.total_size:  # wrap-pad (torch semantics)
                idx += idx[: self.total_size - len(idx)]
        return idx

    def __iter__(self):
        return iter(self._indices()[self.rank::self.num_replicas])

    def __len__(self):
        return self.num_samples

    def set_epoch(self, epoch):
        """Call before each epoch so shuffles differ across epochs while
        staying identical across ranks (torch DistributedSampler API)."""
        self.epoch = int(epoch)


class ElasticSampler(ShardedSampler):
    """ShardedSampler that survives membership changes mid-epoch.

    Usage with axon.elastic: register `sampler.set_world` via
    `state.on_reset` (or call it inside the reset callback) and call
    `sampler.record_batch(batch_size)` after each step; after a restore,
    iteration continues over the not-yet-consumed remainder of the epoch,
    re-sharded over the new world.
    """

    def __init__(self, dataset, shuffle=True, seed=0, num_replicas=None,
                 rank=None):
        self.processed = 0
        super().__init__(dataset, num_replicas=num_replicas, rank=rank,
                         shuffle=shuffle, seed=seed, drop_last=False)

    def set_world(self, num_replicas=None, rank=None):
        self.num_replicas = (num_replicas if num_replicas is not None
                             else basics.size())
        self.rank = rank if rank is not None else basics.rank()
        self._recompute()

    def record_batch(self, global_batch_size):
        """Mark `global_batch_size` samples (whole-job, all ranks) of the
        current epoch as consumed."""
        self.processed = min(self.total_size,
                             self.processed + int(global_batch_size))

    def set_epoch(self, epoch):
        super().set_epoch(epoch)
        self.processed = 0
        self._recompute()

    def _recompute(self):
        n = len(self.dataset)
        self.total_size = math.ceil(n / self.num_replicas) * self.num_replicas
        remaining = max(0, self.total_size - getattr(self, "processed", 0))
        # Remaining work re-sharded over the (possibly new) world; pad so
        # every rank runs the same step count.
        self.num_samples = math.ceil(remaining / self.num_replicas)

    def __iter__(self):
        idx = self._indices()[self.processed:]
        want = self.num_samples * self.num_replicas
        while len(idx) < want and idx:
            idx += idx[: want - len(idx)]
        return iter(idx[self.rank::self.num_replicas])

    def __len__(self):
        return self.num_samples
