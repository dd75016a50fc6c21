"""Worker registry: membership, liveness, model index.

Reimplements the semantics of the reference's WorkerRegistry
(reference: server/src/services/WorkerRegistry.ts):
  * in-memory map mirrored to bus hash "workers" (:238-242)
  * boot re-hydration with staleness filter (:76-110)
  * cleanup sweep removing workers silent > heartbeat_timeout (:182-219)
  * heartbeat-TTL-key probe for fast abrupt-disconnect detection (:125-180)
  * re-registration request on unknown heartbeat (:261-323, :496-515)
  * busy/online flip at max_concurrent_jobs (:421-454)
Differences: tier/resource data is actually populated by workers
(reference gap), and the bus itself announces dropped
worker connections.
"""

from __future__ import annotations

import asyncio
import json
import time
from typing import Callable, Optional

from gridllm_amd.bus.base import Bus
from gridllm_amd.config import ServerConfig
from gridllm_amd.types import Channels, Keys, WorkerInfo, WorkerStatus, now_ms
from gridllm_amd.utils.logging import get_logger

log = get_logger("registry")

_TIER_RANK = {"high": 3, "medium": 2, "low": 1}


class WorkerRegistry:
    def __init__(self, bus: Bus, cfg: Optional[ServerConfig] = None):
        self.bus = bus
        self.cfg = cfg or ServerConfig()
        self.workers: dict[str, WorkerInfo] = {}
        self._tasks: list[asyncio.Task] = []
        self._listeners: dict[str, list[Callable]] = {"worker_removed": [],
                                                      "worker_registered": []}
        self._started = False

    def on(self, event: str, cb: Callable) -> None:
        self._listeners[event].append(cb)

    def _emit(self, event: str, *args) -> None:
        for cb in self._listeners.get(event, []):
            res = cb(*args)
            if asyncio.iscoroutine(res):
                asyncio.get_running_loop().create_task(res)

    # ------------------------------------------------------------- lifecycle

    async def initialize(self) -> None:
        await self.bus.subscribe(Channels.WORKER_REGISTERED, self._h_registered)
        await self.bus.subscribe(Channels.WORKER_UNREGISTERED, self._h_unregistered)
        await self.bus.subscribe(Channels.WORKER_HEARTBEAT, self._h_heartbeat)
        await self.bus.subscribe(Channels.WORKER_STATUS, self._h_status)
        await self.bus.subscribe(Channels.WORKER_DISCONNECTED, self._h_disconnected)
        await self._load_existing()
        loop = asyncio.get_running_loop()
        self._tasks.append(loop.create_task(self._cleanup_loop()))
        self._started = True

    async def shutdown(self) -> None:
        for t in self._tasks:
            t.cancel()
        self._tasks.clear()

    async def _load_existing(self) -> None:
        """Boot re-hydration of workers persisted by a previous server
        (reference: WorkerRegistry.ts:76-110)."""
        stored = await self.bus.hgetall(Keys.WORKERS)
        cutoff = now_ms() - self.cfg.worker_heartbeat_timeout_ms
        for wid, raw in stored.items():
            try:
                info = WorkerInfo.from_dict(json.loads(raw))
            except (json.JSONDecodeError, TypeError):
                await self.bus.hdel(Keys.WORKERS, wid)
                continue
            if info.last_heartbeat >= cutoff:
                self.workers[wid] = info
                log.worker("re-hydrated worker", worker_id=wid)
            else:
                await self.bus.hdel(Keys.WORKERS, wid)

    # --------------------------------------------------------------- events

    async def _h_registered(self, _ch: str, payload: dict) -> None:
        info = WorkerInfo.from_dict(payload)
        if not info.id:
            return
        info.last_heartbeat = now_ms()
        is_new = info.id not in self.workers
        self.workers[info.id] = info
        await self._persist(info)
        log.worker("worker registered", worker_id=info.id, models=info.models,
                   tier=info.performance_tier, tp=info.tp_degree)
        if is_new:
            self._emit("worker_registered", info)

    async def _h_unregistered(self, _ch: str, payload: dict) -> None:
        await self.remove_worker(payload.get("workerId", ""), "unregistered")

    async def _h_disconnected(self, _ch: str, payload: dict) -> None:
        await self.remove_worker(payload.get("workerId", ""), "disconnected")

    async def _h_heartbeat(self, _ch: str, payload: dict) -> None:
        wid = payload.get("workerId", "")
        if not wid:
            return
        info = self.workers.get(wid)
        if info is None:
            # unknown heartbeat: try bus state, else ask to re-register
            raw = await self.bus.hget(Keys.WORKERS, wid)
            if raw:
                try:
                    info = WorkerInfo.from_dict(json.loads(raw))
                    self.workers[wid] = info
                except (json.JSONDecodeError, TypeError):
                    info = None
            if info is None:
                await self.bus.publish(Channels.worker_reregister(wid), {})
                return
        info.last_heartbeat = now_ms()
        if "status" in payload:
            info.status = payload["status"]
        if "currentJobs" in payload:
            info.current_jobs = int(payload["currentJobs"])
        if "gpu" in payload:
            info.gpu = payload["gpu"]
        await self._persist(info)

    async def _h_status(self, _ch: str, payload: dict) -> None:
        wid = payload.get("workerId", "")
        info = self.workers.get(wid)
        if info is None:
            return
        if "status" in payload:
            info.status = payload["status"]
        if "currentJobs" in payload:
            info.current_jobs = int(payload["currentJobs"])
        if "models" in payload:
            info.models = payload["models"]
        await self._persist(info)

    async def _persist(self, info: WorkerInfo) -> None:
        # fire-and-forget: runs on every job-count flip (2x per job) —
        # an awaited RTT here serializes the assignment chain
        await self.bus.hset(Keys.WORKERS, info.id,
                            json.dumps(info.to_dict()), ack=False)

    # --------------------------------------------------------------- sweeps

    async def _cleanup_loop(self) -> None:
        interval = self.cfg.worker_cleanup_interval_ms / 1000.0
        while True:
            await asyncio.sleep(interval)
            try:
                await self._sweep()
            except Exception:  # noqa: BLE001
                log.logger.exception("registry sweep failed")

    async def _sweep(self) -> None:
        cutoff = now_ms() - self.cfg.worker_heartbeat_timeout_ms
        for wid, info in list(self.workers.items()):
            if info.last_heartbeat < cutoff:
                # TTL-key probe before declaring death (ref :161-180)
                hb = await self.bus.kv_get(Keys.heartbeat(wid))
                if hb is None:
                    await self.remove_worker(wid, "heartbeat timeout")

    async def remove_worker(self, wid: str, reason: str) -> None:
        info = self.workers.pop(wid, None)
        if info is None:
            return
        await self.bus.hdel(Keys.WORKERS, wid)
        log.worker("worker removed", worker_id=wid, reason=reason)
        self._emit("worker_removed", wid, reason)

    # --------------------------------------------------------------- queries

    def get(self, wid: str) -> Optional[WorkerInfo]:
        return self.workers.get(wid)

    def all_workers(self) -> list[WorkerInfo]:
        return list(self.workers.values())

    def get_workers_by_model(self, model: str) -> list[WorkerInfo]:
        return [w for w in self.workers.values() if model in w.models]

    def get_available_workers_by_model(self, model: str) -> list[WorkerInfo]:
        """Available = advertises the model, online, has capacity
        (reference: WorkerRegistry.ts:413-419)."""
        return [w for w in self.workers.values()
                if model in w.models
                and w.status not in (WorkerStatus.OFFLINE.value,
                                     WorkerStatus.PAUSED.value)
                and w.current_jobs < w.max_concurrent_jobs]

    def get_all_models(self) -> dict[str, int]:
        """model -> number of workers advertising it (ref :484-494)."""
        out: dict[str, int] = {}
        for w in self.workers.values():
            for m in w.models:
                out[m] = out.get(m, 0) + 1
        return out

    def counts(self) -> dict:
        total = len(self.workers)
        busy = sum(1 for w in self.workers.values()
                   if w.status == WorkerStatus.BUSY.value)
        return {"total": total, "busy": busy, "online": total - busy}

    async def update_job_count(self, wid: str, delta: int) -> None:
        """Flip online<->busy at capacity (reference: :421-454)."""
        info = self.workers.get(wid)
        if info is None:
            return
        info.current_jobs = max(0, info.current_jobs + delta)
        info.status = (WorkerStatus.BUSY.value
                       if info.current_jobs >= info.max_concurrent_jobs
                       else WorkerStatus.ONLINE.value)
        await self._persist(info)

    def is_alive(self, wid: str, within_ms: int = 10_000) -> bool:
        info = self.workers.get(wid)
        return info is not None and (now_ms() - info.last_heartbeat) < within_ms
