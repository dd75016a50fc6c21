"""Env-validated configuration.

Same knob names and defaults as the reference where sensible
(reference: server/src/config/index.ts:6-49, client/src/config/index.ts:6-65),
plus the engine knobs the reference has no analog for (model arch, dtype,
TP degree, KV budget).  Invalid env raises at construction, like the
reference's Joi schema throw (server/src/config/index.ts:45-49).
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Optional


def _env_int(name: str, default: int, lo: int = 0, hi: int = 2**31) -> int:
    raw = os.environ.get(name)
    if raw is None:
        return default
    try:
        v = int(raw)
    except ValueError as e:
        raise ValueError(f"config: {name}={raw!r} is not an integer") from e
    if not (lo <= v <= hi):
        raise ValueError(f"config: {name}={v} out of range [{lo},{hi}]")
    return v


def _env_float(name: str, default: float) -> float:
    raw = os.environ.get(name)
    if raw is None:
        return default
    try:
        return float(raw)
    except ValueError as e:
        raise ValueError(f"config: {name}={raw!r} is not a number") from e


def _env_str(name: str, default: str, choices: Optional[list[str]] = None) -> str:
    v = os.environ.get(name, default)
    if choices and v not in choices:
        raise ValueError(f"config: {name}={v!r} not in {choices}")
    return v


@dataclass
class ServerConfig:
    """Scheduler/gateway process config (reference: server/src/config/index.ts)."""

    host: str = "0.0.0.0"
    port: int = 4000
    bus_host: str = "127.0.0.1"
    bus_port: int = 46379
    worker_heartbeat_timeout_ms: int = 15_000   # ref default 15 s
    worker_cleanup_interval_ms: int = 5_000     # ref default 5 s
    job_timeout_ms: int = 600_000               # ref default 10 min
    job_retry_attempts: int = 3
    job_retry_delay_ms: int = 5_000
    max_concurrent_jobs_per_worker: int = 1     # ref default 1
    status_log_interval_s: int = 60
    log_level: str = "info"
    rate_limit_max_requests: int = 0            # 0 = unlimited (ref: built
    rate_limit_window_ms: int = 900_000         # but unmounted; we mount it)
    bus_external: bool = False                  # connect to a standalone bus
                                                # (python -m gridllm_amd.bus)
                                                # instead of embedding one

    @classmethod
    def from_env(cls) -> "ServerConfig":
        return cls(
            host=_env_str("GRIDLLM_HOST", "0.0.0.0"),
            port=_env_int("PORT", 4000, 1, 65535),
            bus_host=_env_str("BUS_HOST", "127.0.0.1"),
            bus_port=_env_int("BUS_PORT", 46379, 1, 65535),
            worker_heartbeat_timeout_ms=_env_int("WORKER_HEARTBEAT_TIMEOUT", 15_000, 1000),
            worker_cleanup_interval_ms=_env_int("WORKER_CLEANUP_INTERVAL", 5_000, 100),
            job_timeout_ms=_env_int("JOB_TIMEOUT", 600_000, 1000),
            job_retry_attempts=_env_int("JOB_RETRY_ATTEMPTS", 3, 0, 100),
            job_retry_delay_ms=_env_int("JOB_RETRY_DELAY", 5_000, 0),
            max_concurrent_jobs_per_worker=_env_int("MAX_CONCURRENT_JOBS_PER_WORKER", 1, 1, 4096),
            log_level=_env_str("LOG_LEVEL", "info", ["debug", "info", "warn", "error"]),
            rate_limit_max_requests=_env_int("RATE_LIMIT_MAX_REQUESTS", 0, 0),
            rate_limit_window_ms=_env_int("RATE_LIMIT_WINDOW_MS", 900_000, 1000),
            bus_external=_env_int("BUS_EXTERNAL", 0, 0, 1) == 1,
        )


@dataclass
class WorkerConfig:
    """Worker agent process config (reference: client/src/config/index.ts)."""

    worker_id: str = ""                         # empty -> auto  worker-<host>-<dev>
    health_port: int = 3000
    bus_host: str = "127.0.0.1"
    bus_port: int = 46379
    heartbeat_interval_ms: int = 5_000          # ref hardcodes 5 s (config:94)
    resource_monitor_interval_ms: int = 10_000
    stream_flush_ms: int = 25      # min interval between stream publishes
                                   # per job (first token flushes at once)
    max_concurrent_tasks: int = 1
    device: int = 0                             # HIP device index this worker pins
    reconnect_max_attempts: int = 10
    reconnect_base_delay_ms: int = 500
    tp_wedge_timeout_ms: int = 60_000  # TP group: batcher stuck this long
                                       # with work in flight -> exit so the
                                       # launcher fails the group over

    @classmethod
    def from_env(cls) -> "WorkerConfig":
        return cls(
            worker_id=_env_str("WORKER_ID", ""),
            health_port=_env_int("WORKER_PORT", 3000, 1, 65535),
            bus_host=_env_str("BUS_HOST", "127.0.0.1"),
            bus_port=_env_int("BUS_PORT", 46379, 1, 65535),
            heartbeat_interval_ms=_env_int("HEARTBEAT_INTERVAL", 5_000, 200),
            resource_monitor_interval_ms=_env_int("RESOURCE_MONITOR_INTERVAL", 10_000, 500),
            stream_flush_ms=_env_int("STREAM_FLUSH_MS", 25, 0),
            max_concurrent_tasks=_env_int("MAX_CONCURRENT_TASKS", 1, 1, 4096),
            device=_env_int("WORKER_DEVICE", 0, 0, 63),
            tp_wedge_timeout_ms=_env_int("TP_WEDGE_TIMEOUT", 60_000, 1000),
            reconnect_max_attempts=_env_int("RECONNECT_MAX_ATTEMPTS", 10, 1),
            reconnect_base_delay_ms=_env_int("RECONNECT_BASE_DELAY", 500, 10),
        )


@dataclass
class EngineConfig:
    """Inference engine config — no reference analog (engine lived in Ollama)."""

    model: str = "llama3.2:1b"
    dtype: str = "bf16"                        # bf16 | fp16 | fp32
    device: str = "auto"                       # auto | cuda:N | cpu
    max_seq_len: int = 8192
    max_batch_size: int = 64
    kv_cache_budget_mb: int = 0                # 0 -> auto from free HBM
    tp_degree: int = 1
    use_hipgraph: bool = True                  # capture decode step in a hipGraph
    autotune: bool = True                      # per-shape GEMM dispatch tuning
    context_shift: bool = True                 # slide the KV window past max_seq
    weights_path: str = ""                     # safetensors dir; empty -> random init
    seed: int = 0

    @classmethod
    def from_env(cls) -> "EngineConfig":
        return cls(
            model=_env_str("ENGINE_MODEL", "llama3.2:1b"),
            dtype=_env_str("ENGINE_DTYPE", "bf16", ["bf16", "fp16", "fp32"]),
            device=_env_str("ENGINE_DEVICE", "auto"),
            max_seq_len=_env_int("ENGINE_MAX_SEQ_LEN", 8192, 16),
            max_batch_size=_env_int("ENGINE_MAX_BATCH", 64, 1, 4096),
            tp_degree=_env_int("ENGINE_TP", 1, 1, 8),
            use_hipgraph=_env_int("ENGINE_HIPGRAPH", 1, 0, 1) == 1,
            autotune=_env_int("ENGINE_AUTOTUNE", 1, 0, 1) == 1,
            context_shift=_env_int("ENGINE_CONTEXT_SHIFT", 1, 0, 1) == 1,
            weights_path=_env_str("ENGINE_WEIGHTS", ""),
            seed=_env_int("ENGINE_SEED", 0),
        )
