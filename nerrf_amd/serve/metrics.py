"""Prometheus observability for the streaming engine.

The reference plans Prometheus metrics as M2 work (reference ROADMAP.md:59,
docs tracker/overview.mdx:266-268) but ships none; here they are real:
ingest/score/plan/recovery counters and latencies exported on an HTTP
endpoint via prometheus_client.  All metrics are no-ops when
prometheus_client is unavailable, so the engine has no hard dependency.
"""
from __future__ import annotations

from typing import Optional

try:
    from prometheus_client import Counter, Gauge, Histogram, start_http_server

    _HAVE_PROM = True
except ImportError:  # pragma: no cover
    _HAVE_PROM = False


class _Noop:
    def labels(self, *a, **k):
        return self

    def inc(self, *a, **k):
        pass

    def set(self, *a, **k):
        pass

    def observe(self, *a, **k):
        pass


if _HAVE_PROM:
    EVENTS_INGESTED = Counter("nerrf_events_ingested_total", "trace events ingested")
    EVENTS_DROPPED = Counter("nerrf_events_dropped_total", "events dropped (slow consumer)")
    WINDOWS_SCORED = Counter("nerrf_windows_scored_total", "windows scored")
    ALARMS = Counter("nerrf_alarms_total", "attack alarms raised")
    SCORE_LATENCY = Histogram("nerrf_score_latency_seconds", "window scoring latency")
    PLAN_LATENCY = Histogram("nerrf_plan_latency_seconds", "MCTS planning latency")
    RECOVERY_MS = Histogram("nerrf_recovery_duration_ms", "rollback duration (ms)")
    WINDOW_EVENTS = Gauge("nerrf_window_events", "events in the current window")
    STORE_EVICTED = Gauge("nerrf_store_evicted_events", "events evicted from the delta store")
else:  # pragma: no cover
    EVENTS_INGESTED = EVENTS_DROPPED = WINDOWS_SCORED = ALARMS = _Noop()
    SCORE_LATENCY = PLAN_LATENCY = RECOVERY_MS = _Noop()
    WINDOW_EVENTS = STORE_EVICTED = _Noop()


def serve_metrics(port: int = 9464) -> Optional[int]:
    """Start the Prometheus scrape endpoint; returns the port or None."""
    if not _HAVE_PROM:
        return None
    start_http_server(port)
    return port


def instrument_engine(engine) -> None:
    """Wrap a StreamingEngine's score/plan/respond with metric recording."""
    import time

    orig_score = engine.score_window
    orig_plan = engine.plan
    orig_respond = engine.respond

    def score_window(*a, **k):
        t0 = time.perf_counter()
        det = orig_score(*a, **k)
        SCORE_LATENCY.observe(time.perf_counter() - t0)
        WINDOWS_SCORED.inc()
        WINDOW_EVENTS.set(det.window_events)
        STORE_EVICTED.set(engine.store.evicted_events)
        if det.alarm:
            ALARMS.inc()
        return det

    def plan(*a, **k):
        t0 = time.perf_counter()
        res = orig_plan(*a, **k)
        PLAN_LATENCY.observe(time.perf_counter() - t0)
        return res

    def respond(*a, **k):
        res = orig_respond(*a, **k)
        RECOVERY_MS.observe(res.duration_ms)
        return res

    engine.score_window = score_window
    engine.plan = plan
    engine.respond = respond
