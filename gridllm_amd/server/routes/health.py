"""Health/observability endpoints (reference: server/src/routes/health.ts).

/health (:16-28), /health/live (:31-39), /health/ready (:42-73),
/health/system (:76-111), /health/workers (:114-139), /health/jobs (:142-169)
— extended with per-worker GPU telemetry (reference had none).
"""

from __future__ import annotations

import os
import time

import psutil
from fastapi import APIRouter
from fastapi.responses import JSONResponse

from gridllm_amd.server.registry import WorkerRegistry
from gridllm_amd.server.scheduler import JobScheduler

_START = time.time()


def build_router(scheduler: JobScheduler, registry: WorkerRegistry) -> APIRouter:
    r = APIRouter()

    @r.get("/health")
    async def health():
        return JSONResponse({"status": "healthy",
                             "uptime_s": round(time.time() - _START, 1),
                             "timestamp": int(time.time() * 1000)})

    @r.get("/health/live")
    async def live():
        return JSONResponse({"status": "alive"})

    @r.get("/health/ready")
    async def ready():
        counts = registry.counts()
        ok = True  # bus is in-process; registry/scheduler constructed
        return JSONResponse({
            "status": "ready" if ok else "not_ready",
            "checks": {"bus": "ok", "registry": "ok", "scheduler": "ok"},
            "workers": counts},
            status_code=200 if ok else 503)

    @r.get("/health/system")
    async def system():
        vm = psutil.virtual_memory()
        return JSONResponse({
            "status": "healthy",
            "pid": os.getpid(),
            "cpu_percent": psutil.cpu_percent(interval=0.0),
            "memory": {"total_mb": vm.total // 2**20,
                       "used_mb": (vm.total - vm.available) // 2**20,
                       "percent": vm.percent},
            "uptime_s": round(time.time() - _START, 1)})

    @r.get("/health/workers")
    async def workers():
        return JSONResponse({
            "counts": registry.counts(),
            "workers": [w.to_dict() for w in registry.all_workers()]})

    @r.get("/health/jobs")
    async def jobs():
        return JSONResponse(scheduler.queue_stats())

    return r
