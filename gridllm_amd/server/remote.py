"""Detached gateway support: scheduler/registry facades backed by the bus.

The reference runs ONE server process whose routes call the scheduler
in-process (server/src/index.ts:63-83). At MI355X serving rates a single
gateway event loop caps the node (every streamed chunk of every request
crosses it), so this module lets ANY number of gateway replicas front one
scheduler:

  * `RemoteScheduler` — implements the same submit_and_wait /
    submit_streaming / cancel_job / queue_stats surface the routes use,
    but enqueues via the `job:submit` bus channel (scheduler.py
    `_h_submit`) and awaits the per-request `job:result:<id>` channel —
    the scheduler does not care which process subscribed.
    Streaming chunks arrive on the gateway's own StreamServer (direct
    worker->gateway TCP, bus/stream.py) when one is attached, falling
    back to the bus `job:stream:<id>` channel otherwise.
  * `RemoteRegistry` — read-only registry view refreshed from the
    `workers` hash + the scheduler's 1 s `cluster:stats` snapshot.

Run a replica with `python -m gridllm_amd.gateway --port P` against a
reachable bus.
"""

from __future__ import annotations

import asyncio
import json
import time
from typing import Callable, Optional

from gridllm_amd.bus.base import Bus
from gridllm_amd.bus.stream import StreamServer
from gridllm_amd.types import Channels, InferenceRequest, Keys, WorkerInfo
from gridllm_amd.utils.logging import get_logger

log = get_logger("gateway")


class RemoteScheduler:
    """Scheduler facade for a detached gateway replica."""

    def __init__(self, bus: Bus, stream_server: Optional[StreamServer] = None,
                 retry_budget_ms: int = 30_000):
        self.bus = bus
        self.stream_server = stream_server
        self.retry_budget_ms = retry_budget_ms
        self._stats: dict = {}
        self._stats_ts: float = 0.0
        self._listeners: dict[str, list[Callable]] = {}

    # metrics.init_metrics subscribes to scheduler events; a replica has
    # none locally (counters live with the real scheduler)
    def on(self, _event: str, _cb: Callable) -> None:
        pass

    async def _refresh_stats(self) -> None:
        if time.monotonic() - self._stats_ts < 0.5:
            return
        try:
            raw = await self.bus.kv_get(Keys.CLUSTER_STATS)
            if raw:
                self._stats = json.loads(raw)
                self._stats_ts = time.monotonic()
        except Exception:  # noqa: BLE001
            pass

    def queue_stats(self) -> dict:
        # sync (metrics gauges + status routes); served from the snapshot
        return dict(self._stats.get("queue") or {})

    @property
    def active(self) -> dict:
        # /inference/:id/status probes membership; approximate from the
        # snapshot's active id list
        ids = (self._stats.get("queue") or {}).get("active_ids") or []
        return {i: True for i in ids}

    def queue_position(self, job_id: str) -> Optional[int]:
        ids = (self._stats.get("queue") or {}).get("queued_ids") or []
        try:
            return ids.index(job_id)
        except ValueError:
            return None

    async def _submit(self, req: InferenceRequest) -> None:
        await self.bus.publish(Channels.JOB_SUBMIT, {"job": req.to_dict()})

    def _budget_s(self, req: InferenceRequest) -> float:
        return (req.timeout_ms + self.retry_budget_ms) / 1000.0 + 30.0

    async def submit_and_wait(self, req: InferenceRequest) -> dict:
        fut: asyncio.Future = asyncio.get_running_loop().create_future()

        async def _on_result(_ch: str, payload: dict) -> None:
            if not fut.done():
                fut.set_result(payload)

        ch = Channels.job_result(req.id)
        await self.bus.subscribe(ch, _on_result)
        try:
            await self._submit(req)
            return await asyncio.wait_for(fut, timeout=self._budget_s(req))
        finally:
            await self.bus.unsubscribe(ch, _on_result)

    async def submit_streaming(self, req: InferenceRequest,
                               on_chunk: Callable[[dict], None]) -> dict:
        fut: asyncio.Future = asyncio.get_running_loop().create_future()

        async def _on_result(_ch: str, payload: dict) -> None:
            if not fut.done():
                fut.set_result(payload)

        def _on_stream(_ch: str, payload: dict):
            chunk = payload.get("chunk")
            if chunk is not None:
                return on_chunk(chunk)
            return None

        rch = Channels.job_result(req.id)
        sch = Channels.job_stream(req.id)
        await self.bus.subscribe(rch, _on_result)
        direct = self.stream_server is not None
        if direct:
            req.metadata["streamAddr"] = self.stream_server.addr
            self.stream_server.subscribe(sch, _on_stream)
        else:
            await self.bus.subscribe(sch, _on_stream)
        try:
            await self._submit(req)
            return await asyncio.wait_for(fut, timeout=self._budget_s(req))
        finally:
            await self.bus.unsubscribe(rch, _on_result)
            if direct:
                self.stream_server.unsubscribe(sch, _on_stream)
            else:
                await self.bus.unsubscribe(sch, _on_stream)

    async def cancel_job(self, job_id: str) -> bool:
        await self.bus.publish(Channels.JOB_CANCEL, {"jobId": job_id})
        return True


class RemoteRegistry:
    """Read-only registry view for a detached gateway (refreshed from the
    bus `workers` hash; same query surface the routes use)."""

    def __init__(self, bus: Bus, ttl_s: float = 1.0):
        self.bus = bus
        self.ttl_s = ttl_s
        self._workers: dict[str, WorkerInfo] = {}
        self._ts = 0.0
        self._task: Optional[asyncio.Task] = None

    async def start(self) -> None:
        await self.refresh()
        self._task = asyncio.get_running_loop().create_task(self._loop())

    async def stop(self) -> None:
        if self._task:
            self._task.cancel()
            self._task = None

    async def _loop(self) -> None:
        while True:
            await asyncio.sleep(self.ttl_s)
            try:
                await self.refresh()
            except Exception:  # noqa: BLE001
                pass

    async def refresh(self) -> None:
        raw = await self.bus.hgetall(Keys.WORKERS)
        workers = {}
        for wid, blob in raw.items():
            try:
                workers[wid] = WorkerInfo.from_dict(json.loads(blob))
            except (json.JSONDecodeError, TypeError):
                continue
        self._workers = workers
        self._ts = time.monotonic()

    # ---- query surface used by the routes --------------------------------

    def all_workers(self) -> list[WorkerInfo]:
        return list(self._workers.values())

    def get(self, worker_id: str) -> Optional[WorkerInfo]:
        return self._workers.get(worker_id)

    def get_workers_by_model(self, model: str) -> list[WorkerInfo]:
        return [w for w in self._workers.values() if model in w.models]

    def get_available_workers_by_model(self, model: str) -> list[WorkerInfo]:
        # same availability rule as the live registry
        # (server/registry.py:189-196)
        return [w for w in self._workers.values()
                if model in w.models
                and w.status not in ("offline", "paused")
                and w.current_jobs < w.max_concurrent_jobs]

    def get_all_models(self) -> dict[str, int]:
        out: dict[str, int] = {}
        for w in self._workers.values():
            for m in w.models:
                out[m] = out.get(m, 0) + 1
        return out

    def counts(self) -> dict:
        total = len(self._workers)
        busy = sum(1 for w in self._workers.values() if w.status == "busy")
        return {"total": total, "busy": busy, "online": total - busy}
