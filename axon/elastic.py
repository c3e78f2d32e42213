"""Elastic training: survive rank failures with in-memory state recovery
(capability parity: horovod.elastic — State/TorchState commit/restore and
the `run` decorator; SURVEY.md §2.2 "Elastic (torch)", §5.3c).

Single-node model: `axonrun --elastic` keeps a *generation file* describing
the current membership (generation number, world size, rendezvous socket,
worker-id -> rank map). When a rank dies, every surviving rank's background
loop fails fast (the star control plane breaks), the launcher writes a new
generation (shrinking the world or respawning the worker), and the `run`
wrapper on each survivor rolls the model back to the last `state.commit()`,
re-initializes at the new generation, re-syncs from the new rank 0, and
resumes. A respawned worker starts fresh and receives everything via
`state.sync()`.

    state = axon.elastic.TorchState(model=model, optimizer=opt, epoch=0)

    @axon.elastic.run
    def train(state):
        for epoch in range(state.epoch, EPOCHS):
            ...
            state.epoch = epoch
            state.commit()

    train(state)
"""
import copy
import json
import os
import time

import torch

import axon as hvd
from axon import functions, mpi_ops


class State:
    """Commit/rollback container for arbitrary (deepcopy-able) values.
    Attribute access proxies to the tracked values."""

    _RESERVED = ("_values", "_saved", "_reset_callbacks")

    def __init__(self, **kwargs):
        object.__setattr__(self, "_values", dict(kwargs))
        object.__setattr__(self, "_saved", None)
        object.__setattr__(self, "_reset_callbacks", [])
        self.commit()

    def register_reset_callbacks(self, callbacks):
        """hvd.elastic parity: callbacks invoked after the worker
        re-initializes on a membership change (new generation), once the
        rolled-back state has been restored and re-synced — the place to
        re-shard samplers (axon.data.ElasticSampler.set_world), rescale
        learning rates, or rebuild world-size-dependent objects."""
        self._reset_callbacks.extend(callbacks)

    def on_reset(self):
        for cb in self._reset_callbacks:
            cb()

    def __getattr__(self, name):
        values = object.__getattribute__(self, "_values")
        if name in values:
            return values[name]
        raise AttributeError(name)

    def __setattr__(self, name, value):
        if name in self._RESERVED:
            object.__setattr__(self, name, value)
        else:
            self._values[name] = value

    def commit(self):
        """Snapshot current values (in memory) as the rollback point. In an
        elastic run this is also the membership-change boundary: if the
        launcher has published a newer generation (discovery grew or shrank
        the world), commit raises HostsUpdatedInterrupt AFTER saving so the
        run loop re-joins with this commit intact."""
        self._saved = self._snapshot()
        _check_hosts_updated()

    def restore(self):
        """Roll back to the last commit."""
        if self._saved is not None:
            self._restore_from(self._saved)

    def sync(self, root_rank=0):
        """Broadcast the committed state from root so all ranks agree
        (called automatically by `run` after every (re)initialization)."""
        self._sync_values(root_rank)
        self.commit()

    # -- hooks subclasses refine ------------------------------------------
    def _snapshot(self):
        return copy.deepcopy(self._values)

    def _restore_from(self, saved):
        self._values.update(copy.deepcopy(saved))

    def _sync_values(self, root_rank):
        synced = functions.broadcast_object(self._values, root_rank,
                                            name="elastic.state")
        if hvd.rank() != root_rank:
            self._values.update(synced)


class TorchState(State):
    """State that understands torch modules and optimizers: snapshots are
    CPU state_dict copies; sync uses broadcast_parameters /
    broadcast_optimizer_state so large tensors go through the collective
    path instead of pickle."""

    def _split(self):
        mods, opts, plain = {}, {}, {}
        for k, v in self._values.items():
            if isinstance(v, torch.nn.Module):
                mods[k] = v
            elif isinstance(v, torch.optim.Optimizer):
                opts[k] = v
            else:
                plain[k] = v
        return mods, opts, plain

    def _snapshot(self):
        mods, opts, plain = self._split()
        snap = {"plain": copy.deepcopy(plain), "mods": {}, "opts": {}}
        for k, m in mods.items():
            snap["mods"][k] = {n: t.detach().to("cpu", copy=True)
                               for n, t in m.state_dict().items()}
        for k, o in opts.items():
            snap["opts"][k] = copy.deepcopy(
                {kk: vv for kk, vv in o.state_dict().items()})
        return snap

    def _restore_from(self, saved):
        mods, opts, plain = self._split()
        for k in plain:
            if k in saved["plain"]:
                self._values[k] = copy.deepcopy(saved["plain"][k])
        for k, m in mods.items():
            if k in saved["mods"]:
                m.load_state_dict(saved["mods"][k])
        for k, o in opts.items():
            if k in saved["opts"]:
                o.load_state_dict(copy.deepcopy(saved["opts"][k]))

    def _sync_values(self, root_rank):
        mods, opts, plain = self._split()
        synced = functions.broadcast_object(plain, root_rank,
                                            name="elastic.plain")
        if hvd.rank() != root_rank:
            for k in plain:
                self._values[k] = synced[k]
        for k, m in mods.items():
            functions.broadcast_parameters(m.state_dict(), root_rank)
        for k, o in opts.items():
            functions.broadcast_optimizer_state(o, root_rank)


# ----------------------------------------------------------------- driver

def _genfile_path():
    return os.environ.get("AXON_ELASTIC_GENFILE")


def read_generation(path):
    # launcher writes atomically (tmp + rename); transient read races just
    # retry on the next poll
    with open(path) as f:
        return json.load(f)


class _Evicted(SystemExit):
    pass


class HostsUpdatedInterrupt(Exception):
    """Raised from State.commit() when the launcher has published a newer
    generation (e.g. discovery grew the world): the worker re-joins at the
    commit boundary instead of waiting for a failure (hvd.elastic's
    HostsUpdatedInterrupt role)."""


_current_gen = -1   # generation this process is currently running in
_last_gen_check = 0.0


def _check_hosts_updated():
    """Cheap (throttled) probe for a newer generation; called from
    State.commit() so growth interrupts land at commit boundaries."""
    global _last_gen_check
    path = _genfile_path()
    if path is None or _current_gen < 0:
        return
    now = time.time()
    if now - _last_gen_check < 0.2:
        return
    _last_gen_check = now
    try:
        gen = read_generation(path)
    except (OSError, ValueError):
        return
    if gen["gen"] > _current_gen:
        raise HostsUpdatedInterrupt(
            f"generation {gen['gen']} published (size {gen['size']})")


def _wait_for_generation(path, newer_than, timeout):
    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            gen = read_generation(path)
            if gen["gen"] > newer_than:
                return gen
        except (OSError, ValueError):
            pass
        time.sleep(0.2)
    raise RuntimeError(
        f"axon elastic: no new generation after {timeout}s (launcher gone?)")


def _apply_generation(gen):
    wid = os.environ["AXON_WORKER_ID"]
    ranks = gen["ranks"]
    if wid not in ranks:
        raise _Evicted(0)  # shrunk out of the job — clean exit
    os.environ["AXON_RANK"] = str(ranks[wid])
    os.environ["AXON_SIZE"] = str(gen["size"])
    os.environ["AXON_LOCAL_RANK"] = str(ranks[wid])
    os.environ["AXON_CONTROL_SOCKET"] = gen["socket"]


def _is_internal_error():
    try:
        from axon import _core

        return bool(_core.fatal_error()) or not _core.is_initialized()
    except Exception:
        return True


def run(func):
    """Decorator: run `func(state, ...)` with elastic recovery. Requires
    launch via `axonrun --elastic`. On a peer failure the state rolls back
    to its last commit, the worker re-initializes at the launcher's next
    generation, re-syncs, and calls `func` again."""

    def wrapper(state, *args, **kwargs):
        path = _genfile_path()
        if path is None:
            # not an elastic launch: plain single-shot call
            hvd.init()
            state.sync()
            return func(state, *args, **kwargs)
        timeout = float(os.environ.get("AXON_ELASTIC_TIMEOUT", "60"))
        global _current_gen
        known_gen = -1
        first = True
        while True:
            gen = _wait_for_generation(path, known_gen, timeout)
            known_gen = gen["gen"]
            _apply_generation(gen)
            _current_gen = gen["gen"]
            hvd.init()
            state.sync()
            if not first:
                state.on_reset()  # membership changed: notify user hooks
            first = False
            try:
                return func(state, *args, **kwargs)
            except HostsUpdatedInterrupt:
                # discovery changed the world: the state was just
                # committed — leave this world and re-join, no rollback
                hvd.shutdown()
            except RuntimeError:
                if not _is_internal_error():
                    raise
                # peer failure: roll back, drop the dead world, rejoin
                hvd.shutdown()
                state.restore()

    return wrapper
