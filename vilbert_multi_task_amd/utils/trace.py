"""Observability: request-id tracing, per-stage timing, Prometheus metrics.

The reference has none of this (stdout prints + a wall-clock
print per request at worker.py:544,657-658). Here every request carries a
trace_id from HTTP submit -> queue message -> worker batch -> websocket
push; the worker emits one structured JSON log line per batch with stage
timings (features / forward / decode), and exports Prometheus counters when
prometheus_client is importable (it is in this image).
"""

from __future__ import annotations

import json
import sys
import time
import uuid
from contextlib import contextmanager
from typing import Dict, Optional


def new_trace_id() -> str:
    return uuid.uuid4().hex[:16]


def log_json(event: str, **fields) -> None:
    rec = {"ts": round(time.time(), 3), "event": event}
    rec.update(fields)
    print(json.dumps(rec), file=sys.stdout, flush=True)


class RequestTrace:
    """Stage timer: with trace.stage("forward"): ... ; trace.report()."""

    def __init__(self, trace_id: Optional[str] = None):
        self.trace_id = trace_id or new_trace_id()
        self.stages: Dict[str, float] = {}
        self._t0 = time.perf_counter()

    @contextmanager
    def stage(self, name: str):
        t = time.perf_counter()
        try:
            yield
        finally:
            self.stages[name] = self.stages.get(name, 0.0) + (time.perf_counter() - t)

    def total_ms(self) -> float:
        return (time.perf_counter() - self._t0) * 1e3

    def report(self, event: str = "request", **fields) -> None:
        log_json(
            event,
            trace_id=self.trace_id,
            total_ms=round(self.total_ms(), 2),
            stages_ms={k: round(v * 1e3, 2) for k, v in self.stages.items()},
            **fields,
        )


# ---------------------------------------------------------------------------
# Prometheus (optional, cached singletons)
# ---------------------------------------------------------------------------

_METRICS = None


def get_metrics():
    global _METRICS
    if _METRICS is not None:
        return _METRICS
    try:
        from prometheus_client import Counter, Gauge, Histogram

        class M:
            requests_total = Counter(
                "vilbert_requests_total", "served requests", ["task_id", "status"]
            )
            batch_rows = Histogram(
                "vilbert_batch_rows", "rows per inference batch",
                buckets=(1, 2, 4, 8, 16, 32, 64, 128),
            )
            request_latency = Histogram(
                "vilbert_request_latency_seconds", "end-to-end request latency",
                buckets=(0.01, 0.025, 0.05, 0.1, 0.25, 0.5, 1, 2.5, 5),
            )
            queue_depth = Gauge("vilbert_queue_depth", "ready messages in queue")

        _METRICS = M()
    except Exception:  # pragma: no cover

        class _Noop:
            def labels(self, *a, **k):
                return self

            def inc(self, *a):
                pass

            def observe(self, *a):
                pass

            def set(self, *a):
                pass

        class M:  # type: ignore
            requests_total = _Noop()
            batch_rows = _Noop()
            request_latency = _Noop()
            queue_depth = _Noop()

        _METRICS = M()
    return _METRICS


def start_metrics_server(port: int) -> Optional[int]:
    """Start the Prometheus exposition endpoint. Returns the bound port
    (port=0 binds an ephemeral one when the client lib reports it back),
    or None when prometheus_client is unavailable."""
    try:
        from prometheus_client import start_http_server

        res = start_http_server(port)
        if isinstance(res, tuple) and res and hasattr(res[0], "server_port"):
            return int(res[0].server_port)
        return port if port else None
    except Exception:
        return None
