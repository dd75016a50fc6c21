"""Native inference/admin API (reference: server/src/routes/inference.ts).

POST /inference (client-settable priority/timeout, :17-95),
GET /inference/:id/status (:128-167), DELETE /inference/:id (:170-192),
GET /inference/models (:195-250), GET /inference/queue (:253-286).
"""

from __future__ import annotations

from fastapi import APIRouter, HTTPException, Request
from fastapi.responses import JSONResponse

from gridllm_amd.server.registry import WorkerRegistry
from gridllm_amd.server.scheduler import JobScheduler
from gridllm_amd.types import InferenceRequest


def build_router(scheduler: JobScheduler, registry: WorkerRegistry) -> APIRouter:
    r = APIRouter()

    @r.post("/inference")
    async def submit(request: Request):
        body = await request.json()
        model = body.get("model", "")
        prompt = body.get("prompt", "")
        if not isinstance(prompt, str):
            raise HTTPException(status_code=400,
                                detail="prompt must be a string")
        if not model or not prompt:
            raise HTTPException(status_code=400,
                                detail="model and prompt are required")
        if not registry.get_workers_by_model(model):
            raise HTTPException(status_code=404,
                                detail=f"model '{model}' not found")
        priority = body.get("priority", "medium")
        if priority not in ("high", "medium", "low"):
            raise HTTPException(status_code=400, detail="invalid priority")
        timeout_ms = int(body.get("timeout", 300_000))
        if not (1000 <= timeout_ms <= 3_600_000):
            raise HTTPException(status_code=400, detail="invalid timeout")
        req = InferenceRequest(model=model, prompt=prompt,
                               request_type="generate", priority=priority,
                               timeout_ms=timeout_ms,
                               options=body.get("options") or {})
        res = await scheduler.submit_and_wait(req)
        if res.get("error"):
            raise HTTPException(status_code=500, detail=res["error"])
        return JSONResponse({"id": req.id, "response": res.get("response", ""),
                             "worker_id": res.get("workerId", ""),
                             "timings": res.get("timings", {})})

    @r.get("/inference/models")
    async def models():
        return JSONResponse({
            "models": [
                {"name": m, "workers": n,
                 "available_workers":
                     len(registry.get_available_workers_by_model(m))}
                for m, n in sorted(registry.get_all_models().items())]})

    @r.get("/inference/queue")
    async def queue():
        return JSONResponse({
            "queue": scheduler.queue_stats(),
            "workers": registry.counts()})

    @r.get("/inference/{job_id}/status")
    async def status(job_id: str):
        pos = scheduler.queue_position(job_id)
        if pos is not None:
            return JSONResponse({"id": job_id, "status": "queued",
                                 "position": pos})
        if job_id in scheduler.active:
            a = scheduler.active[job_id]
            return JSONResponse({"id": job_id, "status": "processing",
                                 "worker_id": a.worker_id})
        raise HTTPException(status_code=404, detail="job not found")

    @r.delete("/inference/{job_id}")
    async def cancel(job_id: str):
        ok = await scheduler.cancel_job(job_id)
        if not ok:
            raise HTTPException(status_code=404, detail="job not found")
        return JSONResponse({"id": job_id, "cancelled": True})

    return r
