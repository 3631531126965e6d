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
    # lock-wait vs actual work split (reference
    # filter_predicate.go:951-957): a growing lock share means the
    # SerializedNodeFilter gate is the bottleneck, not the allocator
    VERB_LOCK_WAIT = Histogram(
        "vgpu_scheduler_verb_lock_wait_seconds",
        "Time a verb spent waiting on the serialization lock",
        ["verb"])
    VERB_WORK = Histogram(
        "vgpu_scheduler_verb_work_seconds",
        "Time a verb spent working (excluding lock wait)",
        ["verb"])
    PLACEMENT_OUTCOME = Counter(
        "vgpu_scheduler_placement_outcome_total",
        "Topology/placement outcomes of successful filters",
        ["mode", "outcome"])
    _HAVE_PROM = True
except Exception:  # pragma: no cover
    _HAVE_PROM = False


def observe(verb: str, seconds: float, success: bool) -> None:
    if not _HAVE_PROM:
        return
    s = "true" if success else "false"
    VERB_LATENCY.labels(verb, s).observe(seconds)
    VERB_TOTAL.labels(verb, s).inc()


def observe_split(verb: str, lock_wait: float, work: float) -> None:
    if not _HAVE_PROM:
        return
    VERB_LOCK_WAIT.labels(verb).observe(lock_wait)
    VERB_WORK.labels(verb).observe(work)


def observe_placement(mode: str, outcome: str) -> None:
    if not _HAVE_PROM:
        return
    PLACEMENT_OUTCOME.labels(mode or "none", outcome).inc()
