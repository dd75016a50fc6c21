"""Request batcher: coalesces concurrent jobs into one batched decode.

The reference could only run 1 job per worker and each job hit Ollama
serially (client/src/services/WorkerClientService.ts:500-505). Here a worker
can accept many concurrent jobs; the batcher gathers whatever arrived within
a small window (or while the previous wave was running) and decodes them as
ONE batch — the MI355X decode step is weight-read-bound, so batching is
nearly free throughput.
"""

from __future__ import annotations

import queue
import threading
import time
from concurrent.futures import Future
from dataclasses import dataclass, field
from typing import Callable, Optional

from gridllm_amd.engine.engine import InferenceEngine
from gridllm_amd.engine import chat as chat_mod
from gridllm_amd.utils.logging import get_logger

log = get_logger("batcher")


@dataclass
class _Pending:
    kind: str                      # generate | chat | embedding
    prompt: str = ""
    messages: Optional[list] = None
    inputs: Optional[list[str]] = None
    options: dict = field(default_factory=dict)
    on_token: Optional[Callable[[str, bool], None]] = None
    cancel: Optional[threading.Event] = None
    future: Future = field(default_factory=Future)
    extra_stops: Optional[list[str]] = None
    think: bool = False
    ts: float = field(default_factory=time.monotonic)


class Batcher:
    """Continuous batching by default: requests join the live decode batch
    at the next step boundary and finished rows leave it, so one long
    generation never blocks new arrivals (TTFT = prefill + one step).
    The TP lockstep path (worker/tp_runner.TPBatcher) overrides with
    wave mode: every rank must see identical work, so rank 0 broadcasts
    whole waves and runs each to completion."""

    def __init__(self, engine: InferenceEngine, window_ms: float = 4.0,
                 continuous: bool = True, admit_per_step: int = 128,
                 admit_min: int = 8, admit_wait_ms: float = 15.0):
        self.engine = engine
        self.window_s = window_ms / 1000.0
        self.continuous = continuous and engine.tp.size == 1
        # wedge watchdog state: loop threads stamp last_progress each
        # iteration; busy means rows were in flight at the stamp. The
        # worker agent's heartbeat turns (busy and stale) into a process
        # exit under TP — a hung collective must take the whole group
        # down so the scheduler's orphan recovery can requeue its jobs.
        self.last_progress = time.monotonic()
        self.busy = False
        # bound sequential prefills between decode steps: a burst of new
        # prompts must not stall in-flight rows for the whole burst
        self.admit_per_step = max(1, admit_per_step)
        # coalesce trickle arrivals: a single-prompt admission costs a
        # full eager prefill pass (~8 ms of host launches at 3B); while
        # the batch is busy, hold arrivals until admit_min gathered or
        # admit_wait_ms passed (TTFT pays at most that wait)
        self.admit_min = max(1, admit_min)
        self.admit_wait_s = admit_wait_ms / 1000.0
        self.q: "queue.Queue[_Pending]" = queue.Queue()
        self._stop = threading.Event()
        target = self._loop_continuous if self.continuous else self._loop
        self._thread = threading.Thread(target=target, daemon=True,
                                        name="gridllm-batcher")
        self._thread.start()

    def wedged(self, timeout_s: float) -> bool:
        """True when the loop thread had work in flight at its last stamp
        and has not completed an iteration within timeout_s."""
        return (self.busy
                and time.monotonic() - self.last_progress > timeout_s)

    def shutdown(self) -> None:
        self._stop.set()
        self.q.put(_Pending(kind="__stop__"))
        self._thread.join(timeout=5.0)

    # ------------------------------------------------------------ submit API

    def submit_generate(self, prompt: str, options: dict,
                        on_token=None, cancel=None,
                        extra_stops=None) -> Future:
        p = _Pending(kind="generate", prompt=prompt, options=options,
                     on_token=on_token, cancel=cancel, extra_stops=extra_stops)
        self.q.put(p)
        return p.future

    def submit_chat(self, messages: list, options: dict,
                    on_token=None, cancel=None, think: bool = False) -> Future:
        p = _Pending(kind="chat", messages=messages, options=options,
                     on_token=on_token, cancel=cancel, think=think)
        self.q.put(p)
        return p.future

    def submit_embedding(self, inputs: list[str]) -> Future:
        p = _Pending(kind="embedding", inputs=inputs)
        self.q.put(p)
        return p.future

    # ------------------------------------------------ continuous batching

    def _prep(self, p: _Pending) -> tuple[str, dict]:
        """Render the prompt + merge stop sequences for a generate/chat row."""
        if p.kind == "chat":
            fam = chat_mod.family_of(self.engine.spec.name)
            prompt = chat_mod.render_chat(p.messages or [], think=p.think,
                                          family=fam)
            opts = dict(p.options)
            opts["stop"] = list(opts.get("stop") or []) + \
                chat_mod.chat_stops(fam)
        else:
            prompt = p.prompt
            opts = dict(p.options)
            if p.extra_stops:
                opts["stop"] = list(opts.get("stop") or []) + p.extra_stops
        return prompt, opts

    def _loop_continuous(self) -> None:
        max_batch = max(1, self.engine.cfg.max_batch_size)
        eng = self.engine
        live: list[tuple[_Pending, object]] = []   # (pending, engine row)
        inflight = None                            # pipelined step handle

        def prune() -> None:
            nonlocal live
            still = []
            for p, row in live:
                if row.done:
                    if not p.future.done():
                        p.future.set_result(row.result)
                else:
                    still.append((p, row))
            live = still

        def fail_live(e: Exception) -> None:
            nonlocal live
            eng.abort_rows([row for _, row in live])
            eng.flush_deferred()
            for p, _ in live:
                if not p.future.done():
                    p.future.set_exception(e)
            live = []

        def drain_inflight() -> None:
            nonlocal inflight
            eng.flush_admit()
            prune()
            if inflight is not None:
                try:
                    eng.collect_step(inflight)
                except Exception as e:  # noqa: BLE001
                    log.logger.exception("pipelined collect failed")
                    fail_live(e)
                inflight = None
                eng.flush_deferred()
                prune()

        hold: list[_Pending] = []      # coalescing buffer for arrivals
        while True:
            self.last_progress = time.monotonic()   # wedge watchdog stamp
            self.busy = bool(live) or inflight is not None
            # -- admit: block only when idle, otherwise drain without waiting
            arrivals: list[_Pending] = []
            if not live and inflight is None and not hold:
                try:
                    arrivals.append(self.q.get(timeout=0.5))
                except queue.Empty:
                    if self._stop.is_set():
                        break
                    continue
            # one slot spare: a pipelined retire frees one step late
            while (len(live) + len(hold) + len(arrivals) < max_batch
                   and len(hold) + len(arrivals) < self.admit_per_step
                   and eng.model.cache.n_free > len(arrivals)
                   + len(hold) + 1):
                try:
                    arrivals.append(self.q.get_nowait())
                except queue.Empty:
                    break
            stop_now = False
            for p in arrivals:
                if p.kind == "__stop__":
                    stop_now = True
                elif p.kind == "embedding":
                    drain_inflight()   # embeds share slots: settle state
                    try:
                        p.future.set_result(eng.embed(p.inputs or []))
                    except Exception as e:  # noqa: BLE001
                        p.future.set_exception(e)
                else:
                    hold.append(p)
            gens: list[_Pending] = []
            # deep batches coalesce admissions harder: each prefill pass
            # costs ms of host launches on this thread (decode submission
            # pauses for it), so at live>=256 trade ~4x fewer passes for
            # ~tens of ms of extra queueing — invisible against the
            # saturated cohort wait, and the larger varlen prefill is
            # also more MFMA-efficient
            deep = len(live) >= 256
            a_min = 32 if deep else self.admit_min
            a_wait = (4 * self.admit_wait_s) if deep else self.admit_wait_s
            if hold and (not live
                         or len(hold) >= a_min
                         or stop_now or self._stop.is_set()
                         or time.monotonic() - hold[0].ts
                         >= a_wait):
                gens, hold = hold, []
            if gens:
                items = []
                ok_gens = []
                for p in gens:
                    # a malformed request (e.g. junk chat messages) must
                    # fail ITS OWN future — an uncaught render error here
                    # killed the batcher thread and turned the worker into
                    # a heartbeating black hole (probe r02: every later
                    # request hung)
                    try:
                        prompt, opts = self._prep(p)
                    except Exception as e:  # noqa: BLE001
                        if not p.future.done():
                            p.future.set_exception(e)
                        continue
                    ok_gens.append(p)
                    items.append((prompt, opts, p.on_token, p.cancel, None))
                gens = ok_gens
                # async admission (no first-token sync) only while the
                # pipeline is running and every new row will stay
                # pipeline-eligible: no grammar constraint, no penalties
                # (a penalty row forces the whole NEXT step onto the sync
                # path, which cannot resolve a pending async admission)
                def _async_ok(op: dict) -> bool:
                    return not (op.get("format")
                                or op.get("repeat_penalty")
                                or op.get("presence_penalty")
                                or op.get("frequency_penalty"))
                use_async = (inflight is not None
                             and not eng.tp.enabled
                             and all(_async_ok(op) for _, op, *_ in items))
                try:
                    if use_async:
                        new_rows = eng.start_rows_async(items)
                    else:
                        drain_inflight()
                        new_rows = eng.start_rows(items)
                except Exception as e:  # noqa: BLE001
                    log.logger.exception("batched admission failed")
                    for p in gens:
                        p.future.set_exception(e)
                    new_rows = []
                for p, row in zip(gens, new_rows):
                    if row.done:
                        p.future.set_result(row.result)
                    else:
                        live.append((p, row))
            # -- one decode step for everything in flight
            if live:
                rows = [row for _, row in live]
                if eng.pipeline_ok(rows):
                    # pipelined: submit step N, then process step N-1's
                    # tokens on the host while N runs on the GPU
                    try:
                        nxt = eng.submit_step(rows, prev=inflight)
                    except Exception as e:  # noqa: BLE001
                        log.logger.exception("pipelined submit failed")
                        drain_inflight()
                        fail_live(e)
                        nxt = None
                    if inflight is not None:
                        try:
                            eng.collect_step(inflight)
                        except Exception as e:  # noqa: BLE001
                            log.logger.exception("pipelined collect failed")
                            fail_live(e)
                            nxt = None
                    inflight = nxt
                    prune()
                else:
                    eng.flush_admit()   # a pending async admission must
                    drain_inflight()    # resolve before any sync step
                    prune()
                    if live:
                        try:
                            eng.step_rows([row for _, row in live])
                        except Exception as e:  # noqa: BLE001
                            log.logger.exception("decode step failed")
                            # retire the rows or their KV slots leak
                            # permanently and admission wedges
                            fail_live(e)
                        prune()
            elif inflight is not None:
                drain_inflight()
            if stop_now or (self._stop.is_set() and not live
                            and inflight is None):
                drain_inflight()
                for p, _ in live:
                    if not p.future.done():
                        p.future.set_exception(
                            RuntimeError("batcher shut down"))
                break

    # ------------------------------------------------------ wave loop (TP)

    def _loop(self) -> None:
        max_batch = max(1, self.engine.cfg.max_batch_size)
        while not self._stop.is_set():
            try:
                first = self.q.get(timeout=0.5)
            except queue.Empty:
                continue
            if first.kind == "__stop__":
                break
            wave = [first]
            deadline = time.monotonic() + self.window_s
            while len(wave) < max_batch:
                left = deadline - time.monotonic()
                if left <= 0:
                    break
                try:
                    nxt = self.q.get(timeout=left)
                except queue.Empty:
                    break
                if nxt.kind == "__stop__":
                    self._stop.set()
                    break
                wave.append(nxt)
            self._run_wave(wave)

    def _run_wave(self, wave: list[_Pending]) -> None:
        embeds = [p for p in wave if p.kind == "embedding"]
        gens = [p for p in wave if p.kind in ("generate", "chat")]
        for p in embeds:
            try:
                p.future.set_result(self.engine.embed(p.inputs or []))
            except Exception as e:  # noqa: BLE001
                p.future.set_exception(e)
        if not gens:
            return
        reqs, on_tokens, cancels = [], [], []
        ok_gens = []
        for p in gens:
            try:
                prompt, opts = self._prep(p)
            except Exception as e:  # noqa: BLE001 — malformed request:
                if not p.future.done():   # fail it, keep the loop alive
                    p.future.set_exception(e)
                continue
            ok_gens.append(p)
            reqs.append((prompt, opts))
            on_tokens.append(p.on_token)
            cancels.append(p.cancel)
        gens = ok_gens
        if not gens:
            return
        try:
            results = self.engine.generate_batch(reqs, on_tokens=on_tokens,
                                                 cancels=cancels)
            for p, r in zip(gens, results):
                p.future.set_result(r)
        except Exception as e:  # noqa: BLE001
            log.logger.exception("wave failed")
            for p in gens:
                if not p.future.done():
                    p.future.set_exception(e)
