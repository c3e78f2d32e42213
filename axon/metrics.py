"""Engine metrics (SURVEY.md §5.5 observability): per-op submission and
byte counters plus an optional Prometheus exposition endpoint for
production deployments.

Counters are process-local (per rank). `collect()` is dependency-free;
`start_http_server(port)` exports the same numbers in Prometheus format
via prometheus_client (one scrape target per rank — point the scraper at
rank 0 for job-level dashboards, or at every rank for skew hunting).
"""
import threading

_lock = threading.Lock()
_ops = {}      # op kind -> submissions
_bytes = {}    # op kind -> tensor bytes submitted
_lat = {}      # op kind -> [count, sum_seconds, max_seconds]
_steps = 0     # optimizer steps (DistributedOptimizer.step)


def record_op(kind, nbytes):
    with _lock:
        _ops[kind] = _ops.get(kind, 0) + 1
        _bytes[kind] = _bytes.get(kind, 0) + int(nbytes)


def record_latency(kind, seconds):
    """Submit-to-synchronize latency of one completed op (includes the
    caller's wait; recorded by mpi_ops.synchronize)."""
    with _lock:
        e = _lat.setdefault(kind, [0, 0.0, 0.0])
        e[0] += 1
        e[1] += seconds
        e[2] = max(e[2], seconds)


def record_step():
    global _steps
    with _lock:
        _steps += 1


def collect():
    """Snapshot: {"ops": {kind: n}, "bytes": {kind: n},
    "latency": {kind: {"count", "sum", "max"}}, "optimizer_steps": n}."""
    with _lock:
        return {"ops": dict(_ops), "bytes": dict(_bytes),
                "latency": {k: {"count": v[0], "sum": v[1], "max": v[2]}
                            for k, v in _lat.items()},
                "optimizer_steps": _steps}


def reset():
    global _steps
    with _lock:
        _ops.clear()
        _bytes.clear()
        _lat.clear()
        _steps = 0


def start_http_server(port, addr="127.0.0.1"):
    """Expose the counters at http://addr:port/metrics (Prometheus
    format). Returns the (server, thread) pair from prometheus_client."""
    from prometheus_client import start_http_server as _start
    from prometheus_client.core import (CounterMetricFamily, REGISTRY,
                                        GaugeMetricFamily,
                                        SummaryMetricFamily)

    from axon import basics

    class _Collector:
        def collect(self):
            snap = collect()
            ops = CounterMetricFamily(
                "axon_ops", "collective submissions", labels=["op"])
            for k, v in snap["ops"].items():
                ops.add_metric([k], v)
            yield ops
            byt = CounterMetricFamily(
                "axon_bytes", "tensor bytes submitted to collectives",
                labels=["op"])
            for k, v in snap["bytes"].items():
                byt.add_metric([k], v)
            yield byt
            lat = SummaryMetricFamily(
                "axon_op_latency_seconds",
                "submit-to-synchronize latency", labels=["op"])
            for k, v in snap["latency"].items():
                lat.add_metric([k], count_value=v["count"],
                               sum_value=v["sum"])
            yield lat
            lmax = GaugeMetricFamily("axon_op_latency_max_seconds",
                                     "max observed op latency",
                                     labels=["op"])
            for k, v in snap["latency"].items():
                lmax.add_metric([k], v["max"])
            yield lmax
            yield CounterMetricFamily("axon_optimizer_steps",
                                      "DistributedOptimizer steps",
                                      value=snap["optimizer_steps"])
            g = GaugeMetricFamily("axon_world_size", "engine world size")
            g.add_metric([], basics.size() if basics.is_initialized() else 0)
            yield g

    REGISTRY.register(_Collector())
    return _start(port, addr=addr)
