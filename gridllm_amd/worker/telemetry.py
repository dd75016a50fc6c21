"""Per-GPU telemetry for registry reporting.

The reference never populated systemResources/performanceTier
(client/src/types/index.ts:90-96 vs server/src/types/index.ts:2-23), making
tier-aware scheduling a no-op. Here workers report real HBM/device stats.
"""

from __future__ import annotations

import torch

from gridllm_amd.types import GPUStats


def gpu_stats(device: int = 0) -> GPUStats:
    if not torch.cuda.is_available():
        return GPUStats(device=-1, name="cpu")
    try:
        props = torch.cuda.get_device_properties(device)
        free_b, total_b = torch.cuda.mem_get_info(device)
        return GPUStats(
            device=device,
            name=props.name,
            hbm_total_mb=total_b // 2**20,
            hbm_used_mb=(total_b - free_b) // 2**20,
        )
    except Exception:  # noqa: BLE001
        return GPUStats(device=device, name="unknown")


def performance_tier(stats: GPUStats) -> str:
    """Tier from HBM capacity: MI355X-class (>=200 GB) = high."""
    if stats.hbm_total_mb >= 200_000:
        return "high"
    if stats.hbm_total_mb >= 40_000:
        return "medium"
    return "low" if stats.device >= 0 else "low"
