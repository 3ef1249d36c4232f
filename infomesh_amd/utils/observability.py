"""Observability: in-process Prometheus-style metrics + query tracing +
the rocprof workflow notes.

Reference parity: infomesh/observability/metrics.py (Prometheus text
collector with counters/gauges/histograms capped at 1000 observations,
QueryTrace/QuerySpan per query hop; OTel is optional there and absent
here — no OTel wheel in this image). The GPU analogue of per-hop
tracing is the rocprofv3 workflow documented in profiles/.
"""
from __future__ import annotations

import threading
import time
from collections import defaultdict
from dataclasses import dataclass, field

MAX_OBSERVATIONS = 1000
DEFAULT_BUCKETS = (0.001, 0.005, 0.01, 0.05, 0.1, 0.5, 1.0, 5.0)


def _label_key(labels: dict[str, str] | None) -> str:
    if not labels:
        return ""
    return ",".join(f'{k}="{v}"' for k, v in sorted(labels.items()))


class MetricsRegistry:
    def __init__(self):
        self._lock = threading.Lock()
        self._counters: dict[tuple[str, str], float] = defaultdict(float)
        self._gauges: dict[tuple[str, str], float] = {}
        self._observations: dict[tuple[str, str], list[float]] = \
            defaultdict(list)

    def inc(self, name: str, value: float = 1.0,
            labels: dict[str, str] | None = None) -> None:
        with self._lock:
            self._counters[(name, _label_key(labels))] += value

    def set_gauge(self, name: str, value: float,
                  labels: dict[str, str] | None = None) -> None:
        with self._lock:
            self._gauges[(name, _label_key(labels))] = value

    def observe(self, name: str, value: float,
                labels: dict[str, str] | None = None) -> None:
        with self._lock:
            lst = self._observations[(name, _label_key(labels))]
            lst.append(value)
            if len(lst) > MAX_OBSERVATIONS:
                del lst[: len(lst) // 2]

    def render(self) -> str:
        """Prometheus text exposition format."""
        lines: list[str] = []
        with self._lock:
            for (name, lk), v in sorted(self._counters.items()):
                lines.append(f"# TYPE {name} counter")
                lines.append(f"{name}{{{lk}}} {v}" if lk else f"{name} {v}")
            for (name, lk), v in sorted(self._gauges.items()):
                lines.append(f"# TYPE {name} gauge")
                lines.append(f"{name}{{{lk}}} {v}" if lk else f"{name} {v}")
            for (name, lk), obs in sorted(self._observations.items()):
                if not obs:
                    continue
                lines.append(f"# TYPE {name} histogram")
                base = f"{name}_bucket"
                cum = 0
                for b in DEFAULT_BUCKETS:
                    cum = sum(1 for o in obs if o <= b)
                    lb = (lk + "," if lk else "") + f'le="{b}"'
                    lines.append(f"{base}{{{lb}}} {cum}")
                lb_inf = (lk + "," if lk else "") + 'le="+Inf"'
                lines.append(f"{base}{{{lb_inf}}} {len(obs)}")
                sfx = f"{{{lk}}}" if lk else ""
                lines.append(f"{name}_sum{sfx} {sum(obs)}")
                lines.append(f"{name}_count{sfx} {len(obs)}")
        return "\n".join(lines) + "\n"


@dataclass
class QuerySpan:
    name: str
    start: float
    end: float = 0.0

    @property
    def elapsed_ms(self) -> float:
        return ((self.end or time.time()) - self.start) * 1e3


@dataclass
class QueryTrace:
    """Per-query hop trace (reference: metrics.py:212-258). Hops here
    are pipeline phases: encode / broadcast / shard-score / gather /
    fuse / rerank."""
    query: str
    spans: list[QuerySpan] = field(default_factory=list)
    started: float = field(default_factory=time.time)

    def span(self, name: str) -> "_SpanCtx":
        return _SpanCtx(self, name)

    def report(self) -> dict:
        return {"query": self.query,
                "total_ms": round((time.time() - self.started) * 1e3, 2),
                "spans": {s.name: round(s.elapsed_ms, 2)
                          for s in self.spans}}


class _SpanCtx:
    def __init__(self, trace: QueryTrace, name: str):
        self.trace = trace
        self.span = QuerySpan(name, 0.0)

    def __enter__(self):
        self.span.start = time.time()
        return self.span

    def __exit__(self, *exc):
        self.span.end = time.time()
        self.trace.spans.append(self.span)
        return False


def grafana_dashboard_json(title: str = "infomesh-amd") -> dict:
    """Minimal Grafana dashboard skeleton for the exported metrics
    (reference: metrics.py:317-446)."""
    panels = []
    for i, (name, expr) in enumerate([
        ("QPS", "rate(api_requests_total[1m])"),
        ("search p95 latency", "histogram_quantile(0.95, rate(api_request_seconds_bucket[5m]))"),
        ("index documents", "index_documents"),
        ("GPU HBM bytes", "engine_hbm_bytes"),
        ("credit balance", "credit_balance"),
    ]):
        panels.append({"id": i + 1, "title": name, "type": "timeseries",
                       "targets": [{"expr": expr}]})
    return {"title": title, "panels": panels, "schemaVersion": 39}


def alert_rules_yaml() -> str:
    return """groups:
- name: infomesh-amd
  rules:
  - alert: SearchLatencyHigh
    expr: histogram_quantile(0.95, rate(api_request_seconds_bucket[5m])) > 0.5
    for: 5m
  - alert: IndexStalled
    expr: rate(documents_indexed_total[30m]) == 0
    for: 1h
  - alert: HBMExhausted
    expr: engine_hbm_free_fraction < 0.05
    for: 5m
"""


# ---------------------------------------------------------- OTLP export

def trace_to_otlp(trace: QueryTrace, service_name: str = "infomesh-amd",
                  trace_id: str | None = None) -> dict:
    """Encode a QueryTrace as an OTLP/HTTP JSON ExportTraceServiceRequest
    (reference parity: infomesh/observability/metrics.py:24-70 uses the
    OTel SDK + OTLP exporter; here the OTLP JSON wire format is emitted
    directly — no SDK dependency, same collectors accept it)."""
    import os as _os
    tid = trace_id or _os.urandom(16).hex()
    root_sid = _os.urandom(8).hex()
    t0 = trace.started
    t_end = max((s.end or time.time()) for s in trace.spans) \
        if trace.spans else time.time()

    def ns(t: float) -> str:
        return str(int(t * 1e9))

    spans = [{
        "traceId": tid, "spanId": root_sid,
        "name": "search",
        "kind": 2,  # SERVER
        "startTimeUnixNano": ns(t0), "endTimeUnixNano": ns(t_end),
        "attributes": [{"key": "query",
                        "value": {"stringValue": trace.query}}],
    }]
    for s in trace.spans:
        spans.append({
            "traceId": tid, "spanId": _os.urandom(8).hex(),
            "parentSpanId": root_sid,
            "name": s.name, "kind": 1,  # INTERNAL
            "startTimeUnixNano": ns(s.start),
            "endTimeUnixNano": ns(s.end or time.time()),
        })
    return {"resourceSpans": [{
        "resource": {"attributes": [{
            "key": "service.name",
            "value": {"stringValue": service_name}}]},
        "scopeSpans": [{"scope": {"name": "infomesh_amd"},
                        "spans": spans}],
    }]}


class OtlpExporter:
    """Batched OTLP/HTTP JSON trace exporter. Disabled unless an
    endpoint is configured (reference behavior: OTel optional,
    import-guarded)."""

    def __init__(self, endpoint: str = "", service_name: str = "infomesh-amd",
                 batch_size: int = 32, post_fn=None):
        self.endpoint = endpoint.rstrip("/")
        self.service_name = service_name
        self.batch_size = batch_size
        self._buf: list[dict] = []
        self._post = post_fn or self._default_post
        self.exported = 0
        self.errors = 0

    @property
    def enabled(self) -> bool:
        return bool(self.endpoint)

    def export(self, trace: QueryTrace) -> None:
        if not self.enabled:
            return
        self._buf.append(trace_to_otlp(trace, self.service_name))
        if len(self._buf) >= self.batch_size:
            self.flush()

    def flush(self) -> None:
        buf, self._buf = self._buf, []
        for payload in buf:
            try:
                self._post(f"{self.endpoint}/v1/traces", payload)
                self.exported += 1
            except Exception:
                self.errors += 1

    @staticmethod
    def _default_post(url: str, payload: dict) -> None:
        import httpx
        httpx.post(url, json=payload, timeout=5.0).raise_for_status()
