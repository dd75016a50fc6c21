"""Prometheus metrics for the gateway/scheduler.

The reference parsed METRICS_ENABLED but never implemented metrics
(client/src/config/index.ts:47, unused — SURVEY.md §5.5). Here /metrics is
real: job lifecycle counters off the scheduler's event emitter, queue /
fleet gauges off live stats, a TTFT histogram observed at the gateway's
streaming path, and a generated-token counter (the two halves of the
headline metric)."""

from __future__ import annotations

from typing import Optional

from prometheus_client import (CollectorRegistry, Counter, Gauge, Histogram,
                               generate_latest, CONTENT_TYPE_LATEST)

_M: Optional["_Metrics"] = None


class _Metrics:
    def __init__(self, scheduler, worker_registry):
        self.reg = CollectorRegistry()
        self.completed = Counter("gridllm_jobs_completed_total",
                                 "Jobs completed", registry=self.reg)
        self.failed = Counter("gridllm_jobs_failed_total",
                              "Jobs failed (after retries)", registry=self.reg)
        self.orphaned = Counter("gridllm_jobs_orphaned_total",
                                "Jobs requeued after worker loss",
                                registry=self.reg)
        self.tokens = Counter("gridllm_tokens_generated_total",
                              "Output tokens generated", registry=self.reg)
        self.ttft = Histogram(
            "gridllm_ttft_seconds", "Time to first streamed token",
            buckets=(.005, .01, .025, .05, .1, .25, .5, 1, 2.5, 5, 10, 30),
            registry=self.reg)
        self.queue_wait = Histogram(
            "gridllm_queue_wait_seconds", "Enqueue-to-assignment latency "
            "(the reference's floor was its 1 s dispatch tick)",
            buckets=(.001, .005, .01, .05, .1, .5, 1, 5, 30),
            registry=self.reg)
        self.queue_depth = Gauge("gridllm_queue_depth", "Queued jobs",
                                 registry=self.reg)
        self.active_jobs = Gauge("gridllm_active_jobs", "Jobs executing",
                                 registry=self.reg)
        self.workers = Gauge("gridllm_workers", "Workers by status",
                             ["status"], registry=self.reg)
        self.queue_depth.set_function(
            lambda: scheduler.queue_stats().get("queued", 0))
        self.active_jobs.set_function(
            lambda: scheduler.queue_stats().get("active", 0))
        for status in ("online", "busy", "offline"):
            self.workers.labels(status=status).set_function(
                (lambda st: lambda: worker_registry.counts().get(st, 0))(status))
        scheduler.on("job_completed", lambda *a: self.completed.inc())
        scheduler.on("job_failed", lambda *a: self.failed.inc())
        scheduler.on("job_orphaned", lambda *a: self.orphaned.inc())


def init_metrics(scheduler, worker_registry) -> None:
    global _M
    _M = _Metrics(scheduler, worker_registry)


def observe_ttft(seconds: float) -> None:
    if _M is not None:
        _M.ttft.observe(seconds)


def observe_queue_wait(seconds: float) -> None:
    if _M is not None:
        _M.queue_wait.observe(seconds)


def add_tokens(n) -> None:
    if _M is not None and n:
        _M.tokens.inc(int(n))


def render() -> tuple[bytes, str]:
    if _M is None:
        return b"", CONTENT_TYPE_LATEST
    return generate_latest(_M.reg), CONTENT_TYPE_LATEST
