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


# ---------------------------------------------------------------------------
# Distributed tracing (OTLP-shaped), closing the reference telemetry crate's
# trace surface (/root/reference/crates/telemetry/src/lib.rs:15-49: OTLP
# traces + logs + metrics behind env-configured exporters). Offline here:
# spans are buffered and exported as OTLP/JSON ResourceSpans batches to a
# JSONL sink and/or POSTed to an OTLP/HTTP endpoint (collector-compatible
# `/v1/traces` payload; tested against a local HTTP server).
# ---------------------------------------------------------------------------


class Span:
    __slots__ = ("tracer", "name", "trace_id", "span_id", "parent_id",
                 "start_ns", "end_ns", "attributes", "status_ok")

    def __init__(self, tracer, name, trace_id, span_id, parent_id, attributes):
        self.tracer = tracer
        self.name = name
        self.trace_id = trace_id
        self.span_id = span_id
        self.parent_id = parent_id
        self.start_ns = time.time_ns()
        self.end_ns = None
        self.attributes = attributes
        self.status_ok = True

    def set_attribute(self, key, value):
        self.attributes[key] = value

    def end(self):
        if self.end_ns is None:
            self.end_ns = time.time_ns()
            self.tracer._record(self)

    def __enter__(self):
        self.tracer._stack.append(self)
        return self

    def __exit__(self, exc_type, exc, tb):
        if exc_type is not None:
            self.status_ok = False
            self.attributes["exception.type"] = exc_type.__name__
        if self.tracer._stack and self.tracer._stack[-1] is self:
            self.tracer._stack.pop()
        self.end()
        return False


class Tracer:
    """Minimal OTLP-shaped tracer: span ids per the W3C trace-context sizes,
    batches exported in the OTLP/JSON ResourceSpans schema."""

    def __init__(self, service_name: str = "hypha-amd",
                 endpoint: str | None = None, sink_path: str | None = None,
                 flush_every: int = 64):
        import secrets

        self._rand = secrets.token_hex
        self.service_name = service_name
        self.endpoint = endpoint or os.environ.get("HYPHA_OTLP_ENDPOINT")
        self.sink_path = sink_path or os.environ.get("HYPHA_TRACE_FILE")
        self.flush_every = flush_every
        self._lock = threading.Lock()
        self._finished: list = []
        # implicit-parent stack is PER-THREAD: a span entered on one thread
        # must not become the parent of spans started concurrently on another
        self._local = threading.local()
        self.export_errors = 0

    @property
    def _stack(self) -> list:
        st = getattr(self._local, "stack", None)
        if st is None:
            st = self._local.stack = []
        return st

    def start_span(self, name: str, parent: "Span | None" = None, **attrs) -> Span:
        if parent is None and self._stack:
            parent = self._stack[-1]
        trace_id = parent.trace_id if parent else self._rand(16)
        return Span(self, name, trace_id, self._rand(8),
                    parent.span_id if parent else None, dict(attrs))

    def _record(self, span: Span) -> None:
        with self._lock:
            self._finished.append(span)
            n = len(self._finished)
        if n >= self.flush_every:
            self.flush()

    @staticmethod
    def _attr(k, v):
        if isinstance(v, bool):
            val = {"boolValue": v}
        elif isinstance(v, int):
            val = {"intValue": str(v)}
        elif isinstance(v, float):
            val = {"doubleValue": v}
        else:
            val = {"stringValue": str(v)}
        return {"key": k, "value": val}

    def _batch(self, spans) -> dict:
        return {
            "resourceSpans": [{
                "resource": {"attributes": [
                    self._attr("service.name", self.service_name)]},
                "scopeSpans": [{
                    "scope": {"name": "hypha_amd.telemetry"},
                    "spans": [{
                        "traceId": s.trace_id,
                        "spanId": s.span_id,
                        **({"parentSpanId": s.parent_id} if s.parent_id else {}),
                        "name": s.name,
                        "kind": 1,
                        "startTimeUnixNano": str(s.start_ns),
                        "endTimeUnixNano": str(s.end_ns),
                        "attributes": [self._attr(k, v)
                                       for k, v in s.attributes.items()],
                        "status": {"code": 1 if s.status_ok else 2},
                    } for s in spans],
                }],
            }]
        }

    def flush(self) -> int:
        with self._lock:
            spans, self._finished = self._finished, []
        if not spans:
            return 0
        batch = self._batch(spans)
        payload = json.dumps(batch)
        if self.sink_path:
            try:
                with open(self.sink_path, "a") as f:
                    f.write(payload + "\n")
            except OSError:
                self.export_errors += 1
        if self.endpoint:
            try:
                import urllib.request

                req = urllib.request.Request(
                    self.endpoint.rstrip("/") + "/v1/traces",
                    data=payload.encode(),
                    headers={"Content-Type": "application/json"})
                urllib.request.urlopen(req, timeout=5).read()
            except Exception:
                self.export_errors += 1
        return len(spans)


TRACER = Tracer()


def get_tracer() -> Tracer:
    return TRACER
