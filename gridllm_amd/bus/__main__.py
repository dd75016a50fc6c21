"""Standalone durable bus: `python -m gridllm_amd.bus`.

Runs the control-plane bus as its OWN process with periodic snapshots, so
scheduler state (job queue, active jobs, worker hash) survives the death
of every other process — the reference's external-Redis-with-AOF
guarantee (docker-compose.yml:12-30, appendonly yes at :22; the scheduler
reloads at JobScheduler.ts:82-126). Kill -9 the server mid-load, restart
it against this bus, and queued jobs still complete.

Usage:
  python -m gridllm_amd.bus [--host H] [--port P]
      [--snapshot PATH] [--snapshot-interval-ms N]

The server process joins with BUS_EXTERNAL=1 (ServerConfig.bus_external).
"""

from __future__ import annotations

import argparse
import asyncio
import signal

from gridllm_amd.bus.tcp import BusServer
from gridllm_amd.utils.logging import get_logger

log = get_logger("bus")


async def run(args) -> None:
    server = BusServer(args.host, args.port)
    if args.snapshot and server.load_snapshot(args.snapshot):
        log.worker("bus snapshot restored", path=args.snapshot,
                   kv=len(server.kv), hashes=len(server.hashes))
    await server.start()
    print("BUS_READY", flush=True)

    stop = asyncio.Event()
    loop = asyncio.get_running_loop()
    for sig in (signal.SIGTERM, signal.SIGINT):
        try:
            loop.add_signal_handler(sig, stop.set)
        except NotImplementedError:  # pragma: no cover - non-unix
            pass

    async def snapshot_loop():
        while True:
            await asyncio.sleep(args.snapshot_interval_ms / 1000.0)
            try:
                server.save_snapshot(args.snapshot)
            except Exception:  # noqa: BLE001
                log.logger.exception("snapshot failed")

    snap_task = None
    if args.snapshot:
        snap_task = loop.create_task(snapshot_loop())
    await stop.wait()
    if snap_task:
        snap_task.cancel()
    if args.snapshot:
        try:
            server.save_snapshot(args.snapshot)   # final synchronous save
        except Exception:  # noqa: BLE001
            pass
    await server.stop()


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=46379)
    ap.add_argument("--snapshot", default="",
                    help="snapshot file path ('' disables durability)")
    ap.add_argument("--snapshot-interval-ms", type=int, default=1000)
    args = ap.parse_args()
    asyncio.run(run(args))


if __name__ == "__main__":
    main()
