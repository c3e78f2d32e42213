"""axon.metrics: per-op counters + Prometheus exposition endpoint."""
import urllib.request
import uuid

import torch

from conftest import run_distributed


def _body(rank, size):
    import axon as hvd
    from axon import metrics

    metrics.reset()
    hvd.allreduce(torch.ones(100), name="m0")
    hvd.allreduce(torch.ones(50), name="m1")
    hvd.broadcast(torch.ones(10), root_rank=0, name="mb")
    snap = metrics.collect()
    assert snap["ops"]["allreduce"] == 2
    assert snap["bytes"]["allreduce"] == 150 * 4
    assert snap["ops"]["broadcast"] == 1
    lat = snap["latency"]["allreduce"]
    assert lat["count"] == 2 and lat["sum"] > 0
    assert lat["max"] >= lat["sum"] / lat["count"]

    if rank == 0:
        port = 21000 + (uuid.uuid4().int % 9000)
        metrics.start_http_server(port)
        text = urllib.request.urlopen(
            f"http://127.0.0.1:{port}/metrics", timeout=10).read().decode()
        assert 'axon_ops_total{op="allreduce"} 2.0' in text, text[-800:]
        assert 'axon_bytes_total{op="allreduce"} 600.0' in text
        assert 'axon_op_latency_seconds_count{op="allreduce"} 2.0' in text
        assert "axon_world_size 2.0" in text
    hvd.barrier(name="metrics.done")


def test_metrics_counters_and_prometheus():
    run_distributed(_body, 2)


def _step_counter_body(rank, size):
    import torch.nn as nn

    import axon as hvd
    from axon import metrics

    metrics.reset()
    m = nn.Linear(2, 2)
    opt = hvd.DistributedOptimizer(torch.optim.SGD(m.parameters(), lr=0.1),
                                   named_parameters=m.named_parameters())
    (m(torch.ones(1, 2)).sum()).backward()
    opt.step()
    assert metrics.collect()["optimizer_steps"] == 1


def test_optimizer_step_counter():
    run_distributed(_step_counter_body, 1)
