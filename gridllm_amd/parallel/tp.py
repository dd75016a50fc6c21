"""Tensor-parallel context: RCCL process groups over xGMI.

One process per GPU; `torch.distributed` backend "nccl" IS RCCL on ROCm.
The TP group fronts the registry as a single worker:
rank 0 talks to the control plane, all ranks join every collective.

The reference has no collectives at all — this module is
the xGMI data plane the rebuild adds. Decode all-reduce messages are small
(hidden*batch*2B, tens of kB), so latency dominates and we leave algorithm
choice to RCCL per-size; bucket tuning happens in bench, not here.
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Optional

import torch
import torch.distributed as dist


@dataclass
class TPContext:
    rank: int = 0
    size: int = 1
    group: Optional[object] = None
    # control-plane side channel: small per-step host messages (admission
    # batches, cancel flags) go over gloo so they never touch the GPU or
    # the RCCL stream — the compute group stays graph-capturable
    ctrl_group: Optional[object] = None

    @property
    def enabled(self) -> bool:
        return self.size > 1

    @property
    def backend(self) -> str:
        if self.size <= 1:
            return ""
        return dist.get_backend(self.group)

    def all_reduce(self, x: torch.Tensor) -> torch.Tensor:
        """Sum across the TP group, in place. No-op at size 1."""
        if self.size > 1:
            dist.all_reduce(x, op=dist.ReduceOp.SUM, group=self.group)
        return x

    def broadcast(self, x: torch.Tensor, src: int = 0) -> torch.Tensor:
        if self.size > 1:
            src_global = dist.get_global_rank(self.group, src) \
                if self.group is not None else src
            dist.broadcast(x, src=src_global, group=self.group)
        return x

    def broadcast_cpu(self, x: torch.Tensor, src: int = 0) -> torch.Tensor:
        """Broadcast a small HOST tensor over the gloo control group (no
        GPU sync, no RCCL-stream interaction)."""
        if self.size > 1:
            dist.broadcast(x, src=src, group=self.ctrl_group or self.group)
        return x

    def bcast_obj(self, obj=None, src: int = 0):
        """Broadcast one picklable object over the control group."""
        if self.size <= 1:
            return obj
        buf = [obj]
        dist.broadcast_object_list(buf, src=src,
                                   group=self.ctrl_group or self.group)
        return buf[0]

    def barrier(self) -> None:
        if self.size > 1:
            dist.barrier(group=self.group)

    def shard(self, n: int) -> tuple[int, int]:
        """Split a dimension of size n: returns (start, local_size)."""
        assert n % self.size == 0, f"cannot shard {n} over tp={self.size}"
        loc = n // self.size
        return self.rank * loc, loc


def init_tp_from_env(backend: str = "") -> TPContext:
    """Initialize a world-sized TP group from torchrun env vars.

    Uses nccl (=RCCL) on GPU, gloo on CPU (CI multi-process tests).
    """
    if "WORLD_SIZE" not in os.environ or int(os.environ["WORLD_SIZE"]) <= 1:
        return TPContext()
    if not dist.is_initialized():
        if not backend:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        # bounded collectives make TP-group failure ATOMIC (SURVEY.md §7
        # hard-part 5): if any rank dies, the others' collectives time out
        # and abort the whole process -> the registry removes the worker
        # and in-flight jobs orphan-requeue, instead of hanging forever.
        # (RCCL honors this via the process-group watchdog.)
        from datetime import timedelta
        timeout_s = int(os.environ.get("TP_COLLECTIVE_TIMEOUT_S", "120"))
        dist.init_process_group(backend=backend,
                                timeout=timedelta(seconds=timeout_s))
    # gloo control group beside the compute group (same membership): host
    # messages stay off the RCCL stream; on CPU (already gloo) reuse it
    ctrl = None
    if dist.get_backend() != "gloo":
        ctrl = dist.new_group(backend="gloo")
    return TPContext(rank=dist.get_rank(), size=dist.get_world_size(),
                     group=None, ctrl_group=ctrl)
