"""In-process bus: shared core + per-"process" client facades.

Lets the whole stack (scheduler, N workers, gateway) run in one asyncio
loop for tests — the reference had no such harness (its only test needed a
live Redis + Ollama).
"""

from __future__ import annotations

import asyncio
import time
from collections import defaultdict
from typing import Optional

from gridllm_amd.bus.base import Bus, Handler


class MemoryBusCore:
    """Shared state: channels, kv (with expiry), hashes."""

    def __init__(self) -> None:
        self.subs: dict[str, list[tuple["MemoryBus", Handler]]] = defaultdict(list)
        self.kv: dict[str, tuple[str, float]] = {}   # value, expiry (0 = none)
        self.hashes: dict[str, dict[str, str]] = defaultdict(dict)

    def _expired(self, key: str) -> bool:
        v = self.kv.get(key)
        if v is None:
            return True
        if v[1] and v[1] < time.monotonic():
            del self.kv[key]
            return True
        return False

    async def deliver(self, channel: str, payload: dict) -> None:
        for client, handler in list(self.subs.get(channel, [])):
            if client.closed:
                continue
            asyncio.get_running_loop().create_task(
                self._safe(handler, channel, payload))

    @staticmethod
    async def _safe(handler: Handler, channel: str, payload: dict) -> None:
        try:
            await handler(channel, payload)
        except Exception:  # noqa: BLE001 — a bad handler must not kill the bus
            import traceback
            traceback.print_exc()

    def drop_client(self, client: "MemoryBus") -> list[str]:
        """Remove a client's subscriptions; return worker ids to announce."""
        for ch in list(self.subs):
            self.subs[ch] = [(c, h) for c, h in self.subs[ch] if c is not client]
        return list(client._identities)


class MemoryBus(Bus):
    def __init__(self, core: MemoryBusCore):
        self.core = core
        self.closed = False
        self._identities: list[str] = []

    async def connect(self) -> None:
        self.closed = False

    async def close(self) -> None:
        if self.closed:
            return
        self.closed = True
        ids = self.core.drop_client(self)
        for wid in ids:
            await self.core.deliver("worker:disconnected", {"workerId": wid})

    async def identify(self, worker_id: str) -> None:
        self._identities.append(worker_id)

    async def publish(self, channel: str, payload: dict) -> None:
        await self.core.deliver(channel, payload)

    async def subscribe(self, channel: str, handler: Handler) -> None:
        self.core.subs[channel].append((self, handler))

    async def unsubscribe(self, channel: str,
                          handler: Optional[Handler] = None) -> None:
        subs = self.core.subs.get(channel, [])
        if handler is None:
            self.core.subs[channel] = [(c, h) for c, h in subs if c is not self]
        else:
            self.core.subs[channel] = [
                (c, h) for c, h in subs if not (c is self and h is handler)]

    async def kv_set(self, key: str, value: str, ttl_ms: int = 0) -> None:
        exp = time.monotonic() + ttl_ms / 1000.0 if ttl_ms else 0.0
        self.core.kv[key] = (value, exp)

    async def kv_get(self, key: str) -> Optional[str]:
        if self.core._expired(key):
            return None
        return self.core.kv[key][0]

    async def kv_del(self, key: str) -> None:
        self.core.kv.pop(key, None)

    async def hset(self, key: str, field: str, value: str,
                   ack: bool = True) -> None:
        self.core.hashes[key][field] = value

    async def hget(self, key: str, field: str) -> Optional[str]:
        return self.core.hashes.get(key, {}).get(field)

    async def hgetall(self, key: str) -> dict[str, str]:
        return dict(self.core.hashes.get(key, {}))

    async def hdel(self, key: str, field: str) -> None:
        self.core.hashes.get(key, {}).pop(field, None)
