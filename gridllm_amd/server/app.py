"""Server process assembly (reference: server/src/index.ts GridLLMServer).

Boot order preserved (:214-270): bus -> registry -> scheduler -> HTTP.
The bus server is EMBEDDED here (the reference assumed an external Redis).
Includes the 60 s status log (:249-265) and graceful shutdown (:272-301).
"""

from __future__ import annotations

import asyncio
from typing import Optional

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse

from gridllm_amd import __version__
from gridllm_amd.bus.tcp import BusServer, TcpBus
from gridllm_amd.config import ServerConfig
from gridllm_amd.server.registry import WorkerRegistry
from gridllm_amd.server.scheduler import JobScheduler
from gridllm_amd.server.routes import (health as health_routes,
                                       inference as inference_routes,
                                       ollama as ollama_routes,
                                       openai as openai_routes)
from gridllm_amd.utils.logging import get_logger

log = get_logger("server")


def create_app(scheduler: JobScheduler, registry: WorkerRegistry,
               cfg: Optional[ServerConfig] = None) -> FastAPI:
    app = FastAPI(title="GridLLM-AMD", version=__version__)

    if cfg is not None:
        from gridllm_amd.server.middleware import mount_rate_limiter
        mount_rate_limiter(app, cfg.rate_limit_max_requests,
                           cfg.rate_limit_window_ms)

    from gridllm_amd.server import metrics as metrics_mod
    metrics_mod.init_metrics(scheduler, registry)

    @app.get("/metrics")
    async def metrics():
        from fastapi import Response
        body, ctype = metrics_mod.render()
        return Response(content=body, media_type=ctype)

    ollama_r = ollama_routes.build_router(scheduler, registry)
    app.include_router(ollama_r, prefix="/ollama")  # reference mount point
    app.include_router(ollama_r)                    # drop-in Ollama compat
    app.include_router(openai_routes.build_router(scheduler, registry))
    app.include_router(inference_routes.build_router(scheduler, registry))
    app.include_router(health_routes.build_router(scheduler, registry))

    @app.get("/")
    async def root():
        return JSONResponse({
            "name": "GridLLM-AMD", "version": __version__,
            "status": "running",
            "workers": registry.counts(),
            "queue": scheduler.queue_stats(),
            "endpoints": ["/api/generate", "/api/chat", "/api/tags",
                          "/api/embed", "/v1/completions",
                          "/v1/chat/completions", "/v1/models", "/inference",
                          "/health"],
        })

    @app.exception_handler(Exception)
    async def on_error(request: Request, exc: Exception):
        log.logger.exception("unhandled error")
        return JSONResponse(status_code=500,
                            content={"error": {"message": str(exc),
                                               "code": "INTERNAL_ERROR"}})

    return app


class GridLLMServer:
    def __init__(self, cfg: Optional[ServerConfig] = None):
        self.cfg = cfg or ServerConfig.from_env()
        # external bus (python -m gridllm_amd.bus, durable across server
        # death — the reference's external-Redis guarantee,
        # docker-compose.yml:12-30) vs embedded (single-process default)
        self.bus_server = None if self.cfg.bus_external else \
            BusServer(self.cfg.bus_host, self.cfg.bus_port)
        self.bus = TcpBus(self.cfg.bus_host, self.cfg.bus_port)
        self.registry = WorkerRegistry(self.bus, self.cfg)
        self.scheduler = JobScheduler(self.bus, self.registry, self.cfg)
        self.app = create_app(self.scheduler, self.registry, self.cfg)
        self._status_task: Optional[asyncio.Task] = None

    async def start(self) -> None:
        if self.bus_server is not None:
            await self.bus_server.start()
        await self.bus.connect()
        # direct worker->gateway chunk plane for this process's own
        # streaming requests (bus/stream.py)
        from gridllm_amd.bus.stream import StreamServer
        self.stream_server = StreamServer("127.0.0.1", 0)
        await self.stream_server.start()
        self.scheduler.stream_server = self.stream_server
        await self.registry.initialize()
        await self.scheduler.initialize()
        self._status_task = asyncio.get_running_loop().create_task(
            self._status_loop())
        log.worker("GridLLM-AMD server started", port=self.cfg.port,
                   bus_port=self.cfg.bus_port,
                   bus="external" if self.bus_server is None else "embedded")

    async def shutdown(self) -> None:
        if self._status_task:
            self._status_task.cancel()
        await self.scheduler.shutdown()
        await self.registry.shutdown()
        if getattr(self, "stream_server", None) is not None:
            await self.stream_server.stop()
        await self.bus.close()
        if self.bus_server is not None:
            await self.bus_server.stop()
        log.worker("server shut down")

    async def _status_loop(self) -> None:
        while True:
            await asyncio.sleep(self.cfg.status_log_interval_s)
            log.performance("status", workers=self.registry.counts(),
                            queue=self.scheduler.queue_stats())


def main() -> None:
    import uvicorn

    cfg = ServerConfig.from_env()
    server = GridLLMServer(cfg)

    app = server.app

    @app.on_event("startup")
    async def _startup():
        await server.start()

    @app.on_event("shutdown")
    async def _shutdown():
        await server.shutdown()

    uvicorn.run(app, host=cfg.host, port=cfg.port,
                log_level="warning", timeout_keep_alive=120)


if __name__ == "__main__":
    main()
