"""Detached gateway replica: `python -m gridllm_amd.gateway`.

Serves the full HTTP API (Ollama + OpenAI + inference + health) against a
remote scheduler over the bus (server/remote.py). Any number of replicas
can front one scheduler — the horizontal-scaling story the reference's
single Express process (server/src/index.ts) did not have. Token chunks
arrive over this process's own StreamServer (direct worker->gateway TCP).

Env/flags: --port (HTTP), --bus HOST:PORT, --stream-host (address
advertised to workers; defaults to the host part of --bus being local).
"""

from __future__ import annotations

import argparse
import asyncio
from typing import Optional

from gridllm_amd.bus.stream import StreamServer
from gridllm_amd.bus.tcp import TcpBus
from gridllm_amd.config import ServerConfig
from gridllm_amd.server.app import create_app
from gridllm_amd.server.remote import RemoteRegistry, RemoteScheduler
from gridllm_amd.utils.logging import get_logger

log = get_logger("gateway")


class GatewayReplica:
    def __init__(self, cfg: Optional[ServerConfig] = None,
                 stream_host: str = "127.0.0.1"):
        self.cfg = cfg or ServerConfig.from_env()
        self.bus = TcpBus(self.cfg.bus_host, self.cfg.bus_port)
        self.stream_server = StreamServer(stream_host, 0)
        self.scheduler = RemoteScheduler(self.bus, self.stream_server)
        self.registry = RemoteRegistry(self.bus)
        self.app = create_app(self.scheduler, self.registry, self.cfg)
        self._stats_task = None

    async def start(self) -> None:
        # bus reconnect supervisor: a restarted standalone bus must not
        # strand detached replicas (the worker has the same loop in
        # worker/main.py; in-flight requests fail, new ones serve —
        # TcpBus.connect re-subscribes surviving handler channels)
        disconnected = asyncio.Event()
        self.bus.on_disconnect = disconnected.set
        await self.bus.connect()
        await self.stream_server.start()
        await self.registry.start()
        loop = asyncio.get_running_loop()

        async def _stats_loop():
            while True:
                try:
                    await self.scheduler._refresh_stats()
                except Exception:  # noqa: BLE001 — bus down: supervisor
                    pass           # below reconnects; stats resume then
                await asyncio.sleep(1.0)

        async def _reconnect_loop():
            while True:
                await disconnected.wait()
                disconnected.clear()
                log.logger.warning("bus connection lost; reconnecting")
                delay = 0.5
                while True:
                    try:
                        await self.bus.connect()
                        log.worker("bus reconnected")
                        break
                    except OSError:
                        await asyncio.sleep(delay)
                        delay = min(10.0, delay * 2)

        self._reconnect_task = loop.create_task(_reconnect_loop())
        self._stats_task = loop.create_task(_stats_loop())
        log.worker("gateway replica started", port=self.cfg.port,
                   bus=f"{self.cfg.bus_host}:{self.cfg.bus_port}",
                   stream_addr=self.stream_server.addr)

    async def shutdown(self) -> None:
        if getattr(self, "_reconnect_task", None):
            self._reconnect_task.cancel()
        if self._stats_task:
            self._stats_task.cancel()
        await self.registry.stop()
        await self.stream_server.stop()
        await self.bus.close()


def main() -> None:
    import uvicorn

    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int, default=0)
    ap.add_argument("--host", default="")
    ap.add_argument("--bus", default="")
    ap.add_argument("--stream-host", default="127.0.0.1")
    args = ap.parse_args()

    cfg = ServerConfig.from_env()
    if args.port:
        cfg.port = args.port
    if args.host:
        cfg.host = args.host
    if args.bus:
        host, port = args.bus.rsplit(":", 1)
        cfg.bus_host, cfg.bus_port = host, int(port)

    replica = GatewayReplica(cfg, stream_host=args.stream_host)
    app = replica.app

    @app.on_event("startup")
    async def _up():
        await replica.start()
        print("GATEWAY_READY", flush=True)

    @app.on_event("shutdown")
    async def _down():
        await replica.shutdown()

    uvicorn.run(app, host=cfg.host, port=cfg.port,
                log_level="warning", timeout_keep_alive=120)


if __name__ == "__main__":
    main()
