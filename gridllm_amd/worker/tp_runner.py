"""Tensor-parallel worker: a TP group registers as ONE worker.

Launch with torch.distributed.run --nproc-per-node <TP> (one rank per GPU,
RCCL over xGMI). Rank 0 runs the normal WorkerAgent + control-plane
connection; ranks >0 run a follower loop that receives each engine wave by
broadcast and executes it in lockstep so every collective lines up.

No reference analog (the reference had zero parallelism beyond request
sharding — SURVEY.md §2.4); design per SURVEY.md §5.8 / §7 step 6.

Determinism across ranks:
  * logits are bitwise-identical on every rank (all-reduced activations,
    replicated lm_head) so sampling matches — PROVIDED every request has a
    seed; rank 0 injects one into any request that lacks it before
    broadcasting the wave.
  * cancellation is decided by rank 0 and broadcast per decode step
    (engine._sync_cancel) so rows retire identically everywhere.
"""

from __future__ import annotations

import asyncio
import random

import torch.distributed as dist

from gridllm_amd.bus.tcp import TcpBus
from gridllm_amd.config import EngineConfig, WorkerConfig
from gridllm_amd.engine.batcher import Batcher, _Pending
from gridllm_amd.engine.engine import InferenceEngine
from gridllm_amd.engine import chat as chat_mod
from gridllm_amd.parallel.tp import TPContext, init_tp_from_env
from gridllm_amd.utils.logging import get_logger
from gridllm_amd.worker.agent import WorkerAgent

log = get_logger("tp-runner")


def _inject_seeds(reqs: list[tuple[str, dict]]) -> None:
    for _, opts in reqs:
        if opts.get("seed") is None:
            opts["seed"] = random.randint(0, 2**31 - 1)


class TPBatcher(Batcher):
    """Rank-0 batcher: broadcasts each wave to followers before running it."""

    def __init__(self, engine: InferenceEngine, tp: TPContext,
                 window_ms: float = 4.0):
        self.tp = tp
        super().__init__(engine, window_ms)

    def _run_wave(self, wave: list[_Pending]) -> None:
        """Broadcast-then-execute each sub-wave IN ORDER: followers replay
        waves in broadcast order, so rank 0 must run each wave's collectives
        before broadcasting the next."""
        embeds = [p for p in wave if p.kind == "embedding"]
        gens = [p for p in wave if p.kind in ("generate", "chat")]
        if embeds:
            inputs_all = [i for p in embeds for i in (p.inputs or [])]
            dist.broadcast_object_list([("embed", inputs_all)], src=0)
            try:
                vecs = self.engine.embed(inputs_all)
                k = 0
                for p in embeds:
                    n = len(p.inputs or [])
                    p.future.set_result(vecs[k:k + n])
                    k += n
            except Exception as e:  # noqa: BLE001
                for p in embeds:
                    if not p.future.done():
                        p.future.set_exception(e)
        if gens:
            reqs = []
            for p in gens:
                if p.kind == "chat":
                    prompt = chat_mod.render_chat(p.messages or [])
                    opts = dict(p.options)
                    opts["stop"] = list(opts.get("stop") or []) + \
                        list(chat_mod.CHAT_STOPS)
                else:
                    prompt = p.prompt
                    opts = dict(p.options)
                    if p.extra_stops:
                        opts["stop"] = list(opts.get("stop") or []) + \
                            p.extra_stops
                reqs.append((prompt, opts))
            _inject_seeds(reqs)
            dist.broadcast_object_list([("generate", reqs)], src=0)
            # run locally with callbacks; followers run the same wave
            try:
                results = self.engine.generate_batch(
                    reqs, on_tokens=[p.on_token for p in gens],
                    cancels=[p.cancel for p in gens])
                for p, r in zip(gens, results):
                    p.future.set_result(r)
            except Exception as e:  # noqa: BLE001
                for p in gens:
                    if not p.future.done():
                        p.future.set_exception(e)

    def shutdown(self) -> None:
        try:
            dist.broadcast_object_list([("stop", None)], src=0)
        except Exception:  # noqa: BLE001
            pass
        super().shutdown()


def follower_loop(engine: InferenceEngine) -> None:
    """Ranks >0: execute broadcast waves until told to stop."""
    while True:
        obj: list = [None]
        dist.broadcast_object_list(obj, src=0)
        kind, payload = obj[0]
        if kind == "stop":
            return
        try:
            if kind == "generate":
                engine.generate_batch(payload)
            elif kind == "embed":
                engine.embed(payload)
        except Exception:  # noqa: BLE001
            log.logger.exception("follower wave failed")


async def _run_rank0(wcfg: WorkerConfig, engine: InferenceEngine,
                     tp: TPContext) -> None:
    bus = TcpBus(wcfg.bus_host, wcfg.bus_port)
    await bus.connect()
    batcher = TPBatcher(engine, tp)
    agent = WorkerAgent(bus, engine, wcfg, batcher=batcher)
    await agent.start()
    await asyncio.Event().wait()   # serve forever


def main() -> None:
    import os
    import torch
    tp = init_tp_from_env()
    wcfg = WorkerConfig.from_env()
    ecfg = EngineConfig.from_env()
    local_rank = int(os.environ.get("LOCAL_RANK", tp.rank))
    if torch.cuda.is_available():
        ecfg.device = f"cuda:{local_rank}"
    ecfg.tp_degree = tp.size
    engine = InferenceEngine(ecfg, tp=tp)
    if tp.rank == 0:
        wcfg.worker_id = wcfg.worker_id or f"tp-worker-{os.getpid()}"
        asyncio.run(_run_rank0(wcfg, engine, tp))
    else:
        follower_loop(engine)


if __name__ == "__main__":
    main()
