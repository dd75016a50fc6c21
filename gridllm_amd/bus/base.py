"""Control-plane message bus interface.

The reference's entire inter-process protocol is Redis pub/sub + KV
(reference: server/src/services/RedisService.ts,
client/src/services/RedisConnectionManager.ts). This image ships no Redis,
so the rebuild provides its own bus with the same semantics:

  * pub/sub channels (at-most-once, per-request dynamic channels)
  * KV with TTL (heartbeat keys)
  * hashes (worker registry, active jobs)
  * broker-mediated: workers never talk to each other — tensors NEVER
    cross this layer (TP collectives are RCCL over xGMI, parallel/tp.py).

Two implementations: MemoryBus (in-process, tests + single-process deploy)
and TcpBus (asyncio TCP, the real multi-process control plane).
An improvement over the reference: the TCP server detects a dropped worker
connection itself and publishes worker:disconnected (the reference relied
on the dying client to best-effort publish it —
RedisConnectionManager.ts:113-156).
"""

from __future__ import annotations

from abc import ABC, abstractmethod
from typing import Awaitable, Callable, Optional

Handler = Callable[[str, dict], Awaitable[None]]


class Bus(ABC):
    @abstractmethod
    async def connect(self) -> None: ...

    @abstractmethod
    async def close(self) -> None: ...

    @abstractmethod
    async def identify(self, worker_id: str) -> None:
        """Associate this connection with a worker id so the bus can
        announce worker:disconnected if the connection drops."""

    @abstractmethod
    async def publish(self, channel: str, payload: dict) -> None: ...

    @abstractmethod
    async def subscribe(self, channel: str, handler: Handler) -> None: ...

    @abstractmethod
    async def unsubscribe(self, channel: str,
                          handler: Optional[Handler] = None) -> None: ...

    @abstractmethod
    async def kv_set(self, key: str, value: str, ttl_ms: int = 0) -> None: ...

    @abstractmethod
    async def kv_get(self, key: str) -> Optional[str]: ...

    @abstractmethod
    async def kv_del(self, key: str) -> None: ...

    @abstractmethod
    async def hset(self, key: str, field: str, value: str,
               ack: bool = True) -> None: ...

    @abstractmethod
    async def hget(self, key: str, field: str) -> Optional[str]: ...

    @abstractmethod
    async def hgetall(self, key: str) -> dict[str, str]: ...

    @abstractmethod
    async def hdel(self, key: str, field: str) -> None: ...
