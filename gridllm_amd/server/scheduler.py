"""Job scheduler: priority queue, model-aware least-loaded dispatch,
retries, timeouts, orphan recovery, per-request result/stream channels.

Reimplements the semantics of the reference's JobScheduler
(reference: server/src/services/JobScheduler.ts):
  * priority sort high>medium>low (:144-151)
  * candidate = available workers holding the model; least currentJobs,
    tiebreak performance tier (:317-360)
  * assignment with liveness double-check + persisted active_jobs (:376-414)
  * retries <= job_retry_attempts with delay (:477-509)
  * orphan recovery: requeue at queue FRONT with priority promoted to high
    and requeueCount metadata (:219-315, :553-630)
  * submitAndWait / submitStreamingJob via job:result:<id> / job:stream:<id>
    (:666-856)
  * boot recovery of queue + active jobs from the bus (:82-126)
Improvement over the reference: dispatch is EVENT-DRIVEN (on enqueue, on
worker-free, on registration) instead of a 1 s poll tick (:128-135); a slow
sweep remains only for orphan detection and retry release.
"""

from __future__ import annotations

import asyncio
import json
from typing import Callable, Optional

from gridllm_amd.bus.base import Bus
from gridllm_amd.config import ServerConfig
from gridllm_amd.server.registry import WorkerRegistry, _TIER_RANK
from gridllm_amd.types import (Channels, InferenceRequest, JobAssignment, Keys,
                               Priority, now_ms)
from gridllm_amd.utils.logging import get_logger

log = get_logger("scheduler")

_PRI_RANK = {"high": 3, "medium": 2, "low": 1}


class JobScheduler:
    def __init__(self, bus: Bus, registry: WorkerRegistry,
                 cfg: Optional[ServerConfig] = None):
        self.bus = bus
        self.registry = registry
        self.cfg = cfg or ServerConfig()
        self.queue: list[InferenceRequest] = []
        self._queue_dirty = False
        self._persist_task: Optional[asyncio.Task] = None
        self.active: dict[str, JobAssignment] = {}
        self._timeout_tasks: dict[str, asyncio.Task] = {}
        self._retry_tasks: set[asyncio.Task] = set()
        self._tasks: list[asyncio.Task] = []
        self._dispatch_event = asyncio.Event()
        self._listeners: dict[str, list[Callable]] = {
            "job_completed": [], "job_failed": [], "job_orphaned": []}
        self.stats = {"completed": 0, "failed": 0, "orphaned": 0,
                      "timeouts": 0, "retries": 0}
        # direct token-stream plane (bus/stream.py): when set, streaming
        # jobs carry this server's address so workers send chunks straight
        # here instead of double-hopping through the bus
        self.stream_server = None

    def on(self, event: str, cb: Callable) -> None:
        self._listeners[event].append(cb)

    def _emit(self, event: str, *args) -> None:
        for cb in self._listeners.get(event, []):
            res = cb(*args)
            if asyncio.iscoroutine(res):
                asyncio.get_running_loop().create_task(res)

    # ------------------------------------------------------------- lifecycle

    async def initialize(self) -> None:
        await self.bus.subscribe(Channels.JOB_COMPLETED, self._h_completed)
        await self.bus.subscribe(Channels.JOB_FAILED, self._h_failed)
        await self.bus.subscribe(Channels.JOB_TIMEOUT, self._h_timeout)
        # remote-submit plane: detached gateway replicas enqueue/cancel via
        # bus messages instead of in-process calls (server/remote.py); the
        # per-request result/stream channels work unchanged for them
        await self.bus.subscribe(Channels.JOB_SUBMIT, self._h_submit)
        await self.bus.subscribe(Channels.JOB_CANCEL, self._h_cancel)
        self.registry.on("worker_removed", self._on_worker_removed)
        self.registry.on("worker_registered", lambda _info: self.kick())
        await self._load_existing()
        loop = asyncio.get_running_loop()
        self._tasks.append(loop.create_task(self._dispatch_loop()))
        self._tasks.append(loop.create_task(self._sweep_loop()))
        self._tasks.append(loop.create_task(self._stats_loop()))

    async def _h_submit(self, _ch: str, payload: dict) -> None:
        job = payload.get("job")
        if job:
            await self.add_job(InferenceRequest.from_dict(job))

    async def _h_cancel(self, _ch: str, payload: dict) -> None:
        jid = payload.get("jobId")
        if jid:
            await self.cancel_job(jid)

    async def _stats_loop(self) -> None:
        """Publish a cluster-stats snapshot for detached gateways (they
        serve /health, /inference/queue and metrics gauges from it)."""
        while True:
            try:
                await self.bus.kv_set(Keys.CLUSTER_STATS, json.dumps({
                    "queue": self.queue_stats(),
                    "workers": self.registry.counts(),
                    "ts": now_ms(),
                }))
            except Exception:  # noqa: BLE001
                pass
            await asyncio.sleep(1.0)

    async def shutdown(self) -> None:
        for t in self._tasks + list(self._timeout_tasks.values()) + \
                list(self._retry_tasks):
            t.cancel()
        self._tasks.clear()
        self._timeout_tasks.clear()
        if self._persist_task is not None:
            self._persist_task.cancel()
        if self._queue_dirty:            # final synchronous snapshot
            try:
                await self.bus.kv_set(
                    Keys.JOB_QUEUE,
                    json.dumps([r.to_dict() for r in self.queue]))
            except Exception:  # noqa: BLE001
                pass

    async def _load_existing(self) -> None:
        """Crash-resume: re-load queue and active jobs persisted on the bus
        (reference: JobScheduler.ts:82-126)."""
        raw_q = await self.bus.kv_get(Keys.JOB_QUEUE)
        if raw_q:
            try:
                self.queue = [InferenceRequest.from_dict(d)
                              for d in json.loads(raw_q)]
            except (json.JSONDecodeError, TypeError):
                self.queue = []
        actives = await self.bus.hgetall(Keys.ACTIVE_JOBS)
        now = now_ms()
        for jid, raw in actives.items():
            try:
                a = JobAssignment.from_dict(json.loads(raw))
            except (json.JSONDecodeError, TypeError):
                await self.bus.hdel(Keys.ACTIVE_JOBS, jid)
                continue
            if a.assigned_at + a.timeout_ms < now:
                await self.bus.hdel(Keys.ACTIVE_JOBS, jid)
            else:
                self.active[jid] = a
                self._arm_timeout(a)

    # -------------------------------------------------------------- enqueue

    async def _persist_queue(self) -> None:
        """Debounced: the reference re-serialized the whole queue to Redis
        on EVERY enqueue/dequeue (JobScheduler.ts:656) — O(n^2) bytes under
        load. Mark dirty and flush at most every 50 ms; crash-resume only
        needs a recent snapshot (orphan-requeue covers the gap)."""
        self._queue_dirty = True
        if self._persist_task is None or self._persist_task.done():
            self._persist_task = asyncio.get_running_loop().create_task(
                self._flush_queue())

    async def _flush_queue(self) -> None:
        await asyncio.sleep(0.05)
        while self._queue_dirty:
            self._queue_dirty = False
            try:
                await self.bus.kv_set(
                    Keys.JOB_QUEUE,
                    json.dumps([r.to_dict() for r in self.queue]))
            except Exception:  # noqa: BLE001
                log.logger.warning("queue persist failed")
                return

    async def add_job(self, req: InferenceRequest, front: bool = False) -> None:
        if front:
            self.queue.insert(0, req)
        else:
            self.queue.append(req)
        await self._persist_queue()
        self.kick()

    def kick(self) -> None:
        self._dispatch_event.set()

    def queue_position(self, job_id: str) -> Optional[int]:
        for i, r in enumerate(self.queue):
            if r.id == job_id:
                return i
        return None

    # ------------------------------------------------------------- dispatch

    async def _dispatch_loop(self) -> None:
        while True:
            await self._dispatch_event.wait()
            self._dispatch_event.clear()
            try:
                await self._process_queue()
            except Exception:  # noqa: BLE001
                log.logger.exception("dispatch failed")

    async def _sweep_loop(self) -> None:
        """Slow sweep: orphan detection only (dispatch is event-driven)."""
        while True:
            await asyncio.sleep(1.0)
            try:
                await self._check_orphans()
                if self.queue:
                    self.kick()
            except Exception:  # noqa: BLE001
                log.logger.exception("sweep failed")

    async def _process_queue(self) -> None:
        """Assign queued jobs to workers, priority first (stable within)."""
        if not self.queue:
            return
        self.queue.sort(key=lambda r: -_PRI_RANK.get(r.priority, 2))
        assigned: set[str] = set()
        for req in list(self.queue):
            worker = self._select_worker(req)
            if worker is None:
                continue
            if await self._assign(req, worker):
                assigned.add(req.id)
        if assigned:
            # filter by id — add_job may have appended to self.queue while
            # _assign awaited (a snapshot-replace here would drop those jobs)
            self.queue = [r for r in self.queue if r.id not in assigned]
            await self._persist_queue()

    def _select_worker(self, req: InferenceRequest):
        """Least-loaded, tier tiebreak (reference: :317-360). Load = max of
        the registry's view and the scheduler's own active-job count — a
        worker heartbeat carrying a stale (lower) currentJobs must not cause
        over-assignment."""
        sched_load: dict[str, int] = {}
        for a in self.active.values():
            sched_load[a.worker_id] = sched_load.get(a.worker_id, 0) + 1

        def load(w) -> int:
            return max(w.current_jobs, sched_load.get(w.id, 0))

        cands = [w for w in self.registry.get_available_workers_by_model(req.model)
                 if load(w) < w.max_concurrent_jobs]
        if not cands:
            return None
        cands.sort(key=lambda w: (load(w),
                                  -_TIER_RANK.get(w.performance_tier, 2)))
        return cands[0]

    async def _assign(self, req: InferenceRequest, worker) -> bool:
        # liveness double-check (reference: :376-386)
        if not self.registry.is_alive(worker.id, within_ms=10_000):
            return False
        a = JobAssignment(job_id=req.id, worker_id=worker.id,
                          request=req.to_dict(), timeout_ms=req.timeout_ms)
        self.active[req.id] = a
        # fire-and-forget: awaiting this RTT serialized assignment to
        # ~1.5 ms/job (2+ s of ramp at 768-deep bursts); recovery
        # semantics tolerate a lost record (orphan sweep)
        await self.bus.hset(Keys.ACTIVE_JOBS, req.id,
                            json.dumps(a.to_dict()), ack=False)
        await self.registry.update_job_count(worker.id, +1)
        await self.bus.publish(Channels.worker_job(worker.id),
                               {"type": "job_assignment", "job": req.to_dict()})
        self._arm_timeout(a)
        from gridllm_amd.server import metrics
        metrics.observe_queue_wait(max(0, now_ms() - req.created_at) / 1000.0)
        log.job("job assigned", job_id=req.id, worker_id=worker.id,
                model=req.model, priority=req.priority)
        return True

    def _arm_timeout(self, a: JobAssignment) -> None:
        async def _watch():
            delay = max(0.0, (a.assigned_at + a.timeout_ms - now_ms()) / 1000.0)
            await asyncio.sleep(delay)
            await self._on_job_timeout(a.job_id)
        self._timeout_tasks[a.job_id] = \
            asyncio.get_running_loop().create_task(_watch())

    def _disarm_timeout(self, job_id: str) -> None:
        t = self._timeout_tasks.pop(job_id, None)
        if t:
            t.cancel()

    # ------------------------------------------------------------ lifecycle

    async def _finish_active(self, job_id: str) -> Optional[JobAssignment]:
        a = self.active.pop(job_id, None)
        if a is None:
            return None
        self._disarm_timeout(job_id)
        await self.bus.hdel(Keys.ACTIVE_JOBS, job_id)
        await self.registry.update_job_count(a.worker_id, -1)
        self.kick()
        return a

    async def _h_completed(self, _ch: str, payload: dict) -> None:
        jid = payload.get("jobId", "")
        a = await self._finish_active(jid)
        if a:
            self.stats["completed"] += 1
            self._emit("job_completed", jid, payload)

    async def _h_failed(self, _ch: str, payload: dict) -> None:
        jid = payload.get("jobId", "")
        a = await self._finish_active(jid)
        if a is None:
            return
        req = InferenceRequest.from_dict(a.request)
        await self._maybe_retry(req, payload.get("error", "worker failure"))

    async def _h_timeout(self, _ch: str, payload: dict) -> None:
        await self._on_job_timeout(payload.get("jobId", ""))

    async def _on_job_timeout(self, job_id: str) -> None:
        a = self.active.get(job_id)
        if a is None:
            return
        self.stats["timeouts"] += 1
        log.job("job timeout", job_id=job_id, worker_id=a.worker_id)
        # tell the worker to abort compute (the reference published this but
        # its worker ignored it — WorkerClientService.ts:483-491; ours aborts)
        await self.bus.publish(Channels.worker_job(a.worker_id),
                               {"type": "job_cancellation", "jobId": job_id,
                                "reason": "timeout"})
        await self._finish_active(job_id)
        req = InferenceRequest.from_dict(a.request)
        await self._maybe_retry(req, "job timeout")

    async def _maybe_retry(self, req: InferenceRequest, error: str) -> None:
        attempts = int(req.metadata.get("retryCount", 0))
        if attempts < self.cfg.job_retry_attempts:
            req.metadata["retryCount"] = attempts + 1
            self.stats["retries"] += 1
            log.job("job retry scheduled", job_id=req.id,
                    attempt=attempts + 1, error=error)

            async def _later():
                await asyncio.sleep(self.cfg.job_retry_delay_ms / 1000.0)
                await self.add_job(req, front=False)
            t = asyncio.get_running_loop().create_task(_later())
            self._retry_tasks.add(t)
            t.add_done_callback(self._retry_tasks.discard)
        else:
            self.stats["failed"] += 1
            self._emit("job_failed", req.id, error)
            await self.bus.publish(Channels.job_result(req.id),
                                   {"jobId": req.id, "error": error,
                                    "done": True})

    # ---------------------------------------------------------- orphan path

    async def _on_worker_removed(self, wid: str, reason: str) -> None:
        """Requeue a dead worker's jobs at the FRONT, priority promoted
        (reference: :553-630)."""
        orphans = [a for a in self.active.values() if a.worker_id == wid]
        for a in orphans:
            await self._orphan(a, f"worker {reason}")

    async def _check_orphans(self) -> None:
        """Belt-and-braces sweep (reference: :219-257)."""
        for a in list(self.active.values()):
            w = self.registry.get(a.worker_id)
            if w is None or (now_ms() - w.last_heartbeat) > \
                    self.cfg.worker_heartbeat_timeout_ms:
                await self._orphan(a, "worker lost")

    async def _orphan(self, a: JobAssignment, reason: str) -> None:
        if a.job_id not in self.active:
            return
        del self.active[a.job_id]
        self._disarm_timeout(a.job_id)
        await self.bus.hdel(Keys.ACTIVE_JOBS, a.job_id)
        req = InferenceRequest.from_dict(a.request)
        req.priority = Priority.HIGH.value
        req.metadata["orphaned"] = True
        req.metadata["originalWorkerId"] = a.worker_id
        req.metadata["requeueCount"] = int(req.metadata.get("requeueCount", 0)) + 1
        self.stats["orphaned"] += 1
        log.job("job orphaned, requeued at front", job_id=a.job_id,
                reason=reason, requeue_count=req.metadata["requeueCount"])
        self._emit("job_orphaned", a.job_id, reason)
        await self.add_job(req, front=True)

    # ------------------------------------------------------------ await API

    async def submit_and_wait(self, req: InferenceRequest) -> dict:
        """Enqueue and await the result (reference: :666-711)."""
        fut: asyncio.Future = asyncio.get_running_loop().create_future()

        async def _on_result(_ch: str, payload: dict) -> None:
            if not fut.done():
                fut.set_result(payload)

        ch = Channels.job_result(req.id)
        await self.bus.subscribe(ch, _on_result)
        try:
            await self.add_job(req)
            budget = (req.timeout_ms
                      + self.cfg.job_retry_attempts
                      * (req.timeout_ms + self.cfg.job_retry_delay_ms)) / 1000.0
            return await asyncio.wait_for(fut, timeout=budget + 30.0)
        finally:
            await self.bus.unsubscribe(ch, _on_result)

    async def submit_streaming(self, req: InferenceRequest,
                               on_chunk: Callable[[dict], None]) -> dict:
        """Enqueue, stream chunks to on_chunk, return the final result
        (reference: :713-856)."""
        fut: asyncio.Future = asyncio.get_running_loop().create_future()

        async def _on_result(_ch: str, payload: dict) -> None:
            if not fut.done():
                fut.set_result(payload)

        async def _on_stream(_ch: str, payload: dict) -> None:
            chunk = payload.get("chunk")
            if chunk is not None:
                res = on_chunk(chunk)
                if asyncio.iscoroutine(res):
                    await res

        rch, sch = Channels.job_result(req.id), Channels.job_stream(req.id)
        await self.bus.subscribe(rch, _on_result)
        direct = self.stream_server is not None
        if direct:
            # direct worker->gateway chunk plane (bus/stream.py): chunks
            # skip the bus broker entirely
            req.metadata["streamAddr"] = self.stream_server.addr
            self.stream_server.subscribe(sch, _on_stream)
        else:
            await self.bus.subscribe(sch, _on_stream)
        try:
            await self.add_job(req)
            budget = (req.timeout_ms
                      + self.cfg.job_retry_attempts
                      * (req.timeout_ms + self.cfg.job_retry_delay_ms)) / 1000.0
            return await asyncio.wait_for(fut, timeout=budget + 30.0)
        finally:
            await self.bus.unsubscribe(rch, _on_result)
            if direct:
                self.stream_server.unsubscribe(sch, _on_stream)
            else:
                await self.bus.unsubscribe(sch, _on_stream)

    async def cancel_job(self, job_id: str) -> bool:
        """Cancel queued or active job (reference: :874-908)."""
        for i, r in enumerate(self.queue):
            if r.id == job_id:
                del self.queue[i]
                await self._persist_queue()
                await self.bus.publish(Channels.job_result(job_id),
                                       {"jobId": job_id, "error": "cancelled",
                                        "done": True})
                return True
        a = self.active.get(job_id)
        if a:
            await self.bus.publish(Channels.worker_job(a.worker_id),
                                   {"type": "job_cancellation",
                                    "jobId": job_id, "reason": "cancelled"})
            await self._finish_active(job_id)
            await self.bus.publish(Channels.job_result(job_id),
                                   {"jobId": job_id, "error": "cancelled",
                                    "done": True})
            return True
        return False

    # -------------------------------------------------------------- queries

    def queue_stats(self) -> dict:
        by_pri: dict[str, int] = {"high": 0, "medium": 0, "low": 0}
        for r in self.queue:
            by_pri[r.priority] = by_pri.get(r.priority, 0) + 1
        return {"queued": len(self.queue), "active": len(self.active),
                "active_ids": list(self.active)[:100],
                "queued_ids": [r.id for r in self.queue[:100]],
                "by_priority": by_pri, **self.stats}
