"""Shared wire-protocol schema.

Single source of truth for the control-plane message shapes, replacing the
split (and inconsistent) server/client type trees of the reference
(reference: server/src/types/index.ts:1-160, client/src/types/index.ts:1-144).
Unlike the reference, workers here actually populate performance tier and
system resources so tier-aware scheduling is real
(reference gap: client/src/types/index.ts:90-96 omits them).
"""

from __future__ import annotations

import time
import uuid
from dataclasses import dataclass, field, asdict
from enum import Enum
from typing import Any, Callable, Optional


def now_ms() -> int:
    return int(time.time() * 1000)


def new_id() -> str:
    return str(uuid.uuid4())


class Priority(str, Enum):
    HIGH = "high"
    MEDIUM = "medium"
    LOW = "low"

    @property
    def weight(self) -> int:
        return {"high": 3, "medium": 2, "low": 1}[self.value]


class WorkerStatus(str, Enum):
    ONLINE = "online"
    BUSY = "busy"
    OFFLINE = "offline"
    PAUSED = "paused"            # maintenance: registered, not schedulable


class RequestType(str, Enum):
    GENERATE = "generate"
    CHAT = "chat"
    EMBEDDING = "embedding"


@dataclass
class SamplingOptions:
    """Sampler feature surface (reference: server/src/routes/ollama.ts:26-48,
    client/src/services/OllamaService.ts:108-126)."""

    temperature: Optional[float] = None
    top_p: Optional[float] = None
    top_k: Optional[int] = None
    min_p: Optional[float] = None
    seed: Optional[int] = None
    stop: Optional[list[str]] = None
    num_predict: Optional[int] = None
    num_ctx: Optional[int] = None
    repeat_penalty: Optional[float] = None
    presence_penalty: Optional[float] = None
    frequency_penalty: Optional[float] = None
    format: object = None          # "json" or a JSON-schema dict ->
                                   # grammar-constrained decoding

    @classmethod
    def from_dict(cls, d: Optional[dict]) -> "SamplingOptions":
        if not d:
            return cls()
        known = {f for f in cls.__dataclass_fields__}  # type: ignore[attr-defined]
        return cls(**{k: v for k, v in d.items() if k in known})

    _FLOATS = ("temperature", "top_p", "min_p", "repeat_penalty",
               "presence_penalty", "frequency_penalty")
    _INTS = ("top_k", "seed", "num_predict", "num_ctx")

    @classmethod
    def validate(cls, d: Optional[dict]) -> Optional[str]:
        """Type-check a wire options dict; returns an error message or
        None. The gateway turns this into a 400 — Ollama rejects
        wrongly-typed options at unmarshal time, and without this check
        a junk value (temperature: "hot") surfaced as an opaque
        engine-side TypeError in the job error."""
        if d is None:
            return None
        if not isinstance(d, dict):
            return "options must be an object"
        for k in cls._FLOATS:
            v = d.get(k)
            if v is not None and (isinstance(v, bool)
                                  or not isinstance(v, (int, float))):
                return f"option {k!r} must be a number"
        for k in cls._INTS:
            v = d.get(k)
            if v is not None and (isinstance(v, bool)
                                  or not isinstance(v, int)):
                return f"option {k!r} must be an integer"
        stop = d.get("stop")
        if stop is not None and not isinstance(stop, str) and not (
                isinstance(stop, list)
                and all(isinstance(s, str) for s in stop)):
            return "option 'stop' must be a string or list of strings"
        return None

    def to_dict(self) -> dict:
        return {k: v for k, v in asdict(self).items() if v is not None}


@dataclass
class ChatMessage:
    role: str
    content: str
    images: Optional[list[str]] = None
    tool_calls: Optional[list[dict]] = None

    def to_dict(self) -> dict:
        d: dict[str, Any] = {"role": self.role, "content": self.content}
        if self.images:
            d["images"] = self.images
        if self.tool_calls:
            d["tool_calls"] = self.tool_calls
        return d


@dataclass
class InferenceRequest:
    """A unit of schedulable work (reference: server/src/types/index.ts:76-100)."""

    id: str = field(default_factory=new_id)
    model: str = ""
    prompt: str = ""
    messages: Optional[list[dict]] = None
    input: Optional[list[str]] = None  # embedding inputs
    request_type: str = RequestType.GENERATE.value
    stream: bool = False
    priority: str = Priority.MEDIUM.value
    timeout_ms: int = 300_000
    options: dict = field(default_factory=dict)
    metadata: dict = field(default_factory=dict)
    created_at: int = field(default_factory=now_ms)

    def to_dict(self) -> dict:
        # hand-rolled: dataclasses.asdict deep-copies every nested value
        # (measured 1.2 s of the gateway loop under load — each job is
        # re-serialized several times on the assignment path)
        return {
            "id": self.id, "model": self.model, "prompt": self.prompt,
            "messages": self.messages, "input": self.input,
            "request_type": self.request_type, "stream": self.stream,
            "priority": self.priority, "timeout_ms": self.timeout_ms,
            "options": self.options, "metadata": self.metadata,
            "created_at": self.created_at,
        }

    @classmethod
    def from_dict(cls, d: dict) -> "InferenceRequest":
        known = {f for f in cls.__dataclass_fields__}  # type: ignore[attr-defined]
        return cls(**{k: v for k, v in d.items() if k in known})


@dataclass
class Timings:
    """Per-request engine timings in nanoseconds (Ollama wire contract;
    reference zeroes these on its OpenAI path — OllamaService.ts:156-161 —
    we always populate them from the engine)."""

    total_duration: int = 0
    load_duration: int = 0
    prompt_eval_count: int = 0
    prompt_eval_duration: int = 0
    eval_count: int = 0
    eval_duration: int = 0

    def to_dict(self) -> dict:
        return asdict(self)


@dataclass
class InferenceResponse:
    id: str
    model: str
    response: str = ""
    message: Optional[dict] = None          # chat responses
    embeddings: Optional[list[list[float]]] = None
    done: bool = True
    done_reason: str = "stop"
    context: Optional[list[int]] = None
    timings: Timings = field(default_factory=Timings)
    worker_id: str = ""
    error: Optional[str] = None

    def to_dict(self) -> dict:
        d = asdict(self)
        d["timings"] = self.timings.to_dict()
        return d

    @classmethod
    def from_dict(cls, d: dict) -> "InferenceResponse":
        d = dict(d)
        t = d.pop("timings", None) or {}
        known = {f for f in cls.__dataclass_fields__}  # type: ignore[attr-defined]
        resp = cls(**{k: v for k, v in d.items() if k in known})
        resp.timings = Timings(**{k: v for k, v in t.items()
                                  if k in Timings.__dataclass_fields__})  # type: ignore[attr-defined]
        return resp


@dataclass
class StreamChunk:
    """One streamed token/chunk (reference: client/src/types/index.ts:70-74)."""

    id: str
    response: str
    done: bool = False
    done_reason: Optional[str] = None
    timings: Optional[dict] = None

    def to_dict(self) -> dict:
        d: dict[str, Any] = {"id": self.id, "response": self.response, "done": self.done}
        if self.done_reason is not None:
            d["done_reason"] = self.done_reason
        if self.timings is not None:
            d["timings"] = self.timings
        return d


@dataclass
class GPUStats:
    """Per-GPU telemetry a worker reports (reference had none — SURVEY §2.2 C10)."""

    device: int = -1
    name: str = "cpu"
    hbm_total_mb: int = 0
    hbm_used_mb: int = 0
    temperature_c: float = 0.0
    power_w: float = 0.0

    def to_dict(self) -> dict:
        return asdict(self)


@dataclass
class WorkerInfo:
    """Registry entry (reference: server/src/types/index.ts:2-40)."""

    id: str = ""
    status: str = WorkerStatus.ONLINE.value
    models: list[str] = field(default_factory=list)
    max_concurrent_jobs: int = 1
    current_jobs: int = 0
    performance_tier: str = "medium"   # high | medium | low
    gpu: dict = field(default_factory=dict)
    tp_degree: int = 1                 # >1 → this worker fronts a TP group
    registered_at: int = field(default_factory=now_ms)
    last_heartbeat: int = field(default_factory=now_ms)

    def to_dict(self) -> dict:
        return asdict(self)

    @classmethod
    def from_dict(cls, d: dict) -> "WorkerInfo":
        known = {f for f in cls.__dataclass_fields__}  # type: ignore[attr-defined]
        return cls(**{k: v for k, v in d.items() if k in known})


@dataclass
class JobAssignment:
    """Scheduler→worker assignment record (reference: JobScheduler.ts:388-414)."""

    job_id: str
    worker_id: str
    request: dict
    assigned_at: int = field(default_factory=now_ms)
    timeout_ms: int = 300_000

    def to_dict(self) -> dict:
        return asdict(self)

    @classmethod
    def from_dict(cls, d: dict) -> "JobAssignment":
        known = {f for f in cls.__dataclass_fields__}  # type: ignore[attr-defined]
        return cls(**{k: v for k, v in d.items() if k in known})


# ---------------------------------------------------------------------------
# Bus channel / key naming — the wire protocol of SURVEY.md §2.3, preserved.
# ---------------------------------------------------------------------------

class Channels:
    WORKER_REGISTERED = "worker:registered"
    WORKER_UNREGISTERED = "worker:unregistered"
    WORKER_DISCONNECTED = "worker:disconnected"
    WORKER_HEARTBEAT = "worker:heartbeat"
    WORKER_STATUS = "worker:status_update"
    JOB_COMPLETED = "job:completed"
    JOB_FAILED = "job:failed"
    JOB_TIMEOUT = "job:timeout"
    # remote-submit plane (no reference analog: the reference's routes
    # called the scheduler in-process; detached gateway replicas submit
    # over the bus — server/remote.py)
    JOB_SUBMIT = "job:submit"
    JOB_CANCEL = "job:cancel"

    @staticmethod
    def worker_job(worker_id: str) -> str:
        return f"worker:{worker_id}:job"

    @staticmethod
    def worker_reregister(worker_id: str) -> str:
        return f"worker:{worker_id}:reregister"

    BROADCAST = "broadcast"      # fleet-wide control verbs

    @staticmethod
    def worker_control(worker_id: str) -> str:
        return f"worker:{worker_id}:control"

    @staticmethod
    def job_result(job_id: str) -> str:
        return f"job:result:{job_id}"

    @staticmethod
    def job_stream(job_id: str) -> str:
        return f"job:stream:{job_id}"


class Keys:
    WORKERS = "workers"          # hash worker_id -> WorkerInfo json
    ACTIVE_JOBS = "active_jobs"  # hash job_id -> JobAssignment json
    JOB_QUEUE = "job_queue"      # json array of queued InferenceRequest
    CLUSTER_STATS = "cluster:stats"  # scheduler stats snapshot (1 s) for
                                     # detached gateways

    @staticmethod
    def heartbeat(worker_id: str) -> str:
        return f"heartbeat:{worker_id}"


OnChunk = Callable[[dict], None]
