"""Per-GPU telemetry for registry reporting.

The reference never populated systemResources/performanceTier
(client/src/types/index.ts:90-96 vs server/src/types/index.ts:2-23), making
tier-aware scheduling a no-op. Here workers report real HBM/device stats.
"""

from __future__ import annotations

import time

import torch

from gridllm_amd.types import GPUStats

# TTL cache: gpu_stats runs on EVERY worker status publish (2x per job)
# and hipMemGetInfo measured ~1.3 ms against a busy device — 2.2 s of
# the worker loop thread at a 1024-deep admission ramp (round-2 GPU
# profile, lt_worker_loop.prof). Telemetry freshness of 2 s is plenty
# for tier routing and /health.
_CACHE: dict[int, tuple[float, GPUStats]] = {}
_TTL_S = 2.0


def gpu_stats(device: int = 0) -> GPUStats:
    if not torch.cuda.is_available():
        return GPUStats(device=-1, name="cpu")
    hit = _CACHE.get(device)
    now = time.monotonic()
    if hit is not None and now - hit[0] < _TTL_S:
        return hit[1]
    try:
        props = torch.cuda.get_device_properties(device)
        free_b, total_b = torch.cuda.mem_get_info(device)
        st = GPUStats(
            device=device,
            name=props.name,
            hbm_total_mb=total_b // 2**20,
            hbm_used_mb=(total_b - free_b) // 2**20,
        )
    except Exception:  # noqa: BLE001
        st = GPUStats(device=device, name="unknown")
    _CACHE[device] = (now, st)
    return st


def performance_tier(stats: GPUStats) -> str:
    """Tier from HBM capacity: MI355X-class (>=200 GB) = high."""
    if stats.hbm_total_mb >= 200_000:
        return "high"
    if stats.hbm_total_mb >= 40_000:
        return "medium"
    return "low" if stats.device >= 0 else "low"
