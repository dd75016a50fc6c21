"""Worker process entrypoint (reference: client/src/index.ts).

One process per MI355X GPU (pin with WORKER_DEVICE / HIP_VISIBLE_DEVICES);
for TP groups, launch via torch.distributed.run — rank 0 fronts the group
as a single registered worker, other ranks follow broadcasts
(gridllm_amd/worker/tp_runner.py).

Reconnects to the bus with exponential backoff
(reference: WorkerClientService.ts:281-309).
"""

from __future__ import annotations

import asyncio
import os

from gridllm_amd.bus.tcp import TcpBus
from gridllm_amd.config import EngineConfig, WorkerConfig
from gridllm_amd.engine import InferenceEngine
from gridllm_amd.utils.logging import get_logger
from gridllm_amd.worker.agent import WorkerAgent

log = get_logger("worker-main")


def build_health_app(agent: WorkerAgent):
    """Worker-local health/status HTTP app (reference: client/src/index.ts
    :53-91 + client/src/routes/health.ts:8-59), with real GPU telemetry."""
    from fastapi import FastAPI
    app = FastAPI(title="GridLLM-AMD worker")

    @app.get("/")
    async def root():
        info = agent._info()
        return {"name": "GridLLM-AMD worker", "workerId": agent.worker_id,
                "status": info.status, "models": info.models,
                "currentJobs": info.current_jobs}

    @app.get("/health")
    @app.get("/health/live")
    async def health():
        return {"status": "healthy", "workerId": agent.worker_id}

    @app.get("/health/ready")
    async def ready():
        ok = agent._started and not agent.paused
        return {"status": "ready" if ok else "not_ready",
                "paused": agent.paused}

    @app.get("/health/system")
    async def system():
        info = agent._info()
        return {"workerId": agent.worker_id, "gpu": info.gpu,
                "engine": agent.engine.stats()}

    @app.get("/worker/status")
    async def status():
        return agent._info().to_dict()

    return app


async def run_worker(wcfg: WorkerConfig, ecfg: EngineConfig) -> None:
    engine = InferenceEngine(ecfg)
    # one worker can serve several models on the same GPU (288 GB HBM):
    # ENGINE_MODELS="llama3.2:3b,llama3:8b" adds engines beyond ENGINE_MODEL
    extra = []
    more = [m.strip() for m in
            os.environ.get("ENGINE_MODELS", "").split(",") if m.strip()]
    for name in more:
        if name == ecfg.model:
            continue
        import dataclasses
        extra.append(InferenceEngine(dataclasses.replace(ecfg, model=name)))
    disconnected = asyncio.Event()
    bus = TcpBus(wcfg.bus_host, wcfg.bus_port,
                 on_disconnect=lambda: disconnected.set())
    agent = WorkerAgent(bus, engine, wcfg, extra_engines=extra)
    if wcfg.health_port > 0:
        import uvicorn
        ucfg = uvicorn.Config(build_health_app(agent), host="0.0.0.0",
                              port=wcfg.health_port, log_level="error")
        asyncio.get_running_loop().create_task(uvicorn.Server(ucfg).serve())
    attempt = 0
    while True:
        try:
            await bus.connect()
            attempt = 0
            disconnected.clear()
            await agent.start()
            await disconnected.wait()
            log.worker("bus connection lost; reconnecting",
                       worker_id=agent.worker_id)
            for t in agent._tasks:
                t.cancel()
            agent._tasks.clear()
        except (ConnectionError, OSError) as e:
            attempt += 1
            if attempt > wcfg.reconnect_max_attempts:
                log.logger.error("giving up reconnecting to bus")
                raise
            delay = min(30.0, wcfg.reconnect_base_delay_ms / 1000.0
                        * (2 ** attempt))
            log.worker("reconnect backoff", attempt=attempt, delay_s=delay,
                       error=str(e))
            await asyncio.sleep(delay)


def main() -> None:
    wcfg = WorkerConfig.from_env()
    ecfg = EngineConfig.from_env()
    if "WORKER_DEVICE" in os.environ:
        ecfg.device = f"cuda:{wcfg.device}"
    asyncio.run(run_worker(wcfg, ecfg))


if __name__ == "__main__":
    main()
