"""Telemetry: metrics registry, bandwidth accounting, optional Prometheus
exposition and JSONL export.

Parity with the reference's telemetry crate: the
reference wires OTLP traces/logs/metrics plus a libp2p bandwidth-counting
transport wrapper (telemetry/src/bandwidth.rs). This environment has no OTLP
collector, so the same signals are exposed through prometheus_client (when a
port is configured) and a JSONL sink; comm byte counters hook the RCCL comm
wrapper the same way the reference wraps its transport.
"""

from __future__ import annotations

import json
import os
import threading
import time
from collections import defaultdict


class Metrics:
    """Process-wide metric registry: counters + gauges + simple timers."""

    def __init__(self):
        self._lock = threading.Lock()
        self.counters: dict[str, float] = defaultdict(float)
        self.gauges: dict[str, float] = {}
        self._sink_path: str | None = None
        self._prom_started = False

    def counter_add(self, name: str, value: float = 1.0, **labels) -> None:
        with self._lock:
            self.counters[self._key(name, labels)] += value

    def gauge_set(self, name: str, value: float, **labels) -> None:
        with self._lock:
            self.gauges[self._key(name, labels)] = value

    @staticmethod
    def _key(name, labels):
        if not labels:
            return name
        tag = ",".join(f"{k}={v}" for k, v in sorted(labels.items()))
        return f"{name}{{{tag}}}"

    def snapshot(self) -> dict:
        with self._lock:
            return {"counters": dict(self.counters), "gauges": dict(self.gauges),
                    "ts": time.time()}

    # -- exporters ----------------------------------------------------------

    def enable_jsonl(self, path: str, interval_s: float = 10.0) -> None:
        self._sink_path = path

        def loop():
            while True:
                time.sleep(interval_s)
                try:
                    with open(path, "a") as f:
                        f.write(json.dumps(self.snapshot()) + "\n")
                except OSError:
                    pass

        threading.Thread(target=loop, daemon=True).start()

    def enable_prometheus(self, port: int) -> bool:
        """Expose /metrics via prometheus_client if available."""
        try:
            import prometheus_client as prom
        except ImportError:
            return False
        if self._prom_started:
            return True

        class Collector:
            def collect(inner):
                from prometheus_client.core import GaugeMetricFamily

                snap = self.snapshot()
                for k, v in {**snap["counters"], **snap["gauges"]}.items():
                    name = k.split("{")[0].replace(".", "_").replace("-", "_")
                    g = GaugeMetricFamily(name, k)
                    g.add_metric([], v)
                    yield g

        prom.REGISTRY.register(Collector())
        prom.start_http_server(port)
        self._prom_started = True
        return True


METRICS = Metrics()


def collect_transport_bytes() -> dict | None:
    """Pull the C++ control-plane transport byte counters (net.cpp
    bandwidth_stats — every framed request/stream byte) into the registry.
    Returns the raw stats, or None when the native core is not built."""
    try:
        from hypha_amd import _core
    except ImportError:
        return None
    s = _core.bandwidth_stats()
    METRICS.gauge_set("hypha.bandwidth.transport.inbound_bytes", s["inbound_bytes"])
    METRICS.gauge_set("hypha.bandwidth.transport.outbound_bytes", s["outbound_bytes"])
    return s


def instrument_comm(comm) -> None:
    """Wrap a parallel.Comm so collective payload bytes are counted —
    the analogue of the reference's bandwidth::Transport wrapper
    (telemetry/src/bandwidth.rs:33-60)."""
    orig = comm.all_reduce_mean_flat

    def wrapped(flat, *a, **kw):
        METRICS.counter_add(
            "hypha.bandwidth.collective.payload_bytes",
            flat.numel() * flat.element_size(),
        )
        METRICS.counter_add("hypha.collectives.count")
        return orig(flat, *a, **kw)

    comm.all_reduce_mean_flat = wrapped
