"""Prometheus metrics for the scheduler extender (reference
pkg/scheduler/metrics): per-verb latency histograms + outcome counters.
"""
from __future__ import annotations

try:
    from prometheus_client import Counter, Histogram

    VERB_LATENCY = Histogram(
        "vgpu_scheduler_verb_duration_seconds",
        "Latency of scheduler extender verbs",
        ["verb", "success"])
    VERB_TOTAL = Counter(
        "vgpu_scheduler_verb_total",
        "Scheduler extender verb invocations",
        ["verb", "success"])
    _HAVE_PROM = True
except Exception:  # pragma: no cover
    _HAVE_PROM = False


def observe(verb: str, seconds: float, success: bool) -> None:
    if not _HAVE_PROM:
        return
    s = "true" if success else "false"
    VERB_LATENCY.labels(verb, s).observe(seconds)
    VERB_TOTAL.labels(verb, s).inc()
