"""Token sampling.

Implements the full option surface the gateway accepts
(reference contract: server/src/routes/ollama.ts:26-48 and
client/src/services/OllamaService.ts:108-126): temperature, top_k, top_p,
min_p, seed, repeat/presence/frequency penalties, stop sequences
(stop handling lives in the engine loop; this module picks token ids).

Runs on-device with torch ops (rocBLAS-free elementwise/sort paths);
greedy is a pure argmax. Per-request RNG via a seeded torch.Generator.
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Optional

import torch

from gridllm_amd.types import SamplingOptions


@dataclass
class SamplerState:
    opts: SamplingOptions
    gen: Optional[torch.Generator]
    device: torch.device
    # token -> count for penalties (SPARSE: a generation sees few unique
    # tokens; dense [vocab] tensors per row cost memory and per-row GPU
    # launches at serving batch sizes)
    counts: Optional[dict] = None
    # counter-based RNG stream for the BATCHED sampler: per-row noise is a
    # pure function of (noise_seed, draws), so the whole batch draws with
    # ONE tensor expression instead of a torch.rand launch per row
    noise_seed: int = 0
    draws: int = 0

    @classmethod
    def create(cls, opts: SamplingOptions, device: torch.device) -> "SamplerState":
        gen = None
        if opts.seed is not None:
            gen = torch.Generator(device=device)
            gen.manual_seed(int(opts.seed))
            noise_seed = int(opts.seed)
        else:
            noise_seed = int(torch.randint(0, 2**62, (1,)).item())
        return cls(opts=opts, gen=gen, device=device, noise_seed=noise_seed)

    @property
    def greedy(self) -> bool:
        t = self.opts.temperature
        return t is not None and t <= 0.0

    @property
    def need_filter(self) -> bool:
        o = self.opts
        return bool((o.top_k and o.top_k > 0)
                    or (o.top_p and 0.0 < o.top_p < 1.0)
                    or (o.min_p and 0.0 < o.min_p < 1.0))

    @property
    def has_penalty(self) -> bool:
        o = self.opts
        return bool(o.repeat_penalty or o.presence_penalty
                    or o.frequency_penalty)

    def observe(self, token_id: int, vocab: int) -> None:
        self.draws += 1
        o = self.opts
        if (o.repeat_penalty or o.presence_penalty or o.frequency_penalty):
            if self.counts is None:
                self.counts = {}
            self.counts[token_id] = self.counts.get(token_id, 0) + 1


def sample(logits: torch.Tensor, state: SamplerState) -> int:
    """Pick one token id from logits [vocab] (f32)."""
    o = state.opts
    if state.counts:
        toks = torch.tensor(list(state.counts), dtype=torch.long,
                            device=logits.device)
        cnts = torch.tensor([float(c) for c in state.counts.values()],
                            device=logits.device)
        vals = logits[toks]
        if o.repeat_penalty and o.repeat_penalty != 1.0:
            vals = torch.where(vals > 0, vals / o.repeat_penalty,
                               vals * o.repeat_penalty)
        if o.presence_penalty:
            vals = vals - o.presence_penalty
        if o.frequency_penalty:
            vals = vals - o.frequency_penalty * cnts
        logits = logits.clone()
        logits[toks] = vals

    temp = o.temperature if o.temperature is not None else 0.8
    if temp <= 0.0:
        return int(torch.argmax(logits).item())
    logits = logits / temp

    if o.top_k and o.top_k > 0 and o.top_k < logits.shape[-1]:
        kth = torch.topk(logits, o.top_k).values[-1]
        logits = torch.where(logits < kth, torch.full_like(logits, float("-inf")),
                             logits)
    probs = torch.softmax(logits, dim=-1)

    if o.min_p and 0.0 < o.min_p < 1.0:
        cutoff = probs.max() * o.min_p
        probs = torch.where(probs < cutoff, torch.zeros_like(probs), probs)

    if o.top_p and 0.0 < o.top_p < 1.0:
        sp, si = torch.sort(probs, descending=True)
        cum = torch.cumsum(sp, dim=-1)
        keep = cum - sp < o.top_p          # keep tokens until mass reaches top_p
        sp = torch.where(keep, sp, torch.zeros_like(sp))
        idx = torch.multinomial(sp, 1, generator=state.gen)
        return int(si[idx].item())

    s = probs.sum()
    if not torch.isfinite(s) or s <= 0:
        return int(torch.argmax(logits).item())
    return int(torch.multinomial(probs, 1, generator=state.gen).item())


class BatchPlan:
    """Per-batch-composition cache for `sample_batch_vectorized`.

    The per-row tensors (temperatures, filter thresholds, greedy mask,
    hash seeds) are constant for a row's whole life, but the batch
    composition only changes on admission/retire — at serving load that
    is every few steps, while the OLD code rebuilt ~8 host->device
    tensors from Python list comprehensions EVERY step (measured 4.5 s
    of 14 s in the worker's decode thread at c=192; on GPU each rebuild
    is also an H2D copy + launch outside the hipGraph). The engine owns
    one plan per live batch and the cache invalidates by object identity
    of the state list. Determinism is unchanged: noise keys stay a pure
    function of (seed, draw index), recomputed vectorized each step."""

    __slots__ = ("states", "temps", "greedy_mask", "all_greedy",
                 "need_filter", "kidx", "mps", "tps", "any_penalty",
                 "seeds_dev", "draws_dev", "nongreedy_dev", "_draws0")

    def __init__(self):
        self.states: list = []

    def refresh(self, states: list["SamplerState"], dev, v: int) -> None:
        import numpy as np
        if len(states) == len(self.states) and \
                all(a is b for a, b in zip(states, self.states)):
            # any_penalty can flip on mid-life (first penalized token
            # observed), so re-derive the cheap flag every step
            self.any_penalty = any(s.counts for s in states)
            return
        self.states = list(states)
        b = len(states)
        # ONE packed H2D per composition change instead of ~8 per-field
        # tensor() calls; the per-step key math runs fully ON DEVICE
        # (draws_dev += 1 after each draw) — the per-step host->device key
        # upload measured 1.7 ms/step of GPU-sync stalls at c192 (round-2
        # GPU profile, lt_worker_batcher.prof)
        host = np.zeros((6, b), dtype=np.float64)
        g = np.zeros(b, dtype=bool)
        seeds = np.zeros(b, dtype=np.int64)
        draws = np.zeros(b, dtype=np.int64)
        for i, s in enumerate(states):
            o = s.opts
            host[0, i] = max(o.temperature if o.temperature is not None
                             else 0.8, 1e-4)
            g[i] = s.greedy
            host[1, i] = 0.0 if s.greedy else 1.0
            host[2, i] = (int(o.top_k) - 1
                          if o.top_k and 0 < o.top_k < v else v - 1)
            host[3, i] = (o.min_p if o.min_p and 0 < o.min_p < 1 else 0.0)
            host[4, i] = (o.top_p if o.top_p and 0 < o.top_p < 1 else 1.0)
            # mask to 63 bits: the device-side key math runs in int64
            seeds[i] = s.noise_seed & ((1 << 63) - 1)
            draws[i] = s.draws
        self.all_greedy = bool(g.all())
        packed = torch.from_numpy(host).to(dev, non_blocking=False)
        self.temps = packed[0].float().unsqueeze(1)
        self.greedy_mask = packed[1].float().unsqueeze(1)
        self.need_filter = any(
            (s.opts.top_k and 0 < s.opts.top_k < v)
            or (s.opts.top_p and 0.0 < s.opts.top_p < 1.0)
            or (s.opts.min_p and 0.0 < s.opts.min_p < 1.0)
            for s in states)
        if self.need_filter:
            self.kidx = packed[2].long().unsqueeze(1)
            self.mps = packed[3].float().unsqueeze(1)
            self.tps = packed[4].float().unsqueeze(1)
        ints = torch.from_numpy(np.stack([seeds, draws])).to(dev)
        self.seeds_dev = ints[0]
        self.draws_dev = ints[1]
        self._draws0 = draws.copy()          # host mirror for validation
        self.nongreedy_dev = packed[1].to(torch.int64)
        self.any_penalty = any(s.counts for s in states)

    def keys(self, states: list["SamplerState"], dev) -> torch.Tensor:
        """Per-row splitmix keys for this step's draw — computed ON DEVICE
        from the cached (seed, draw-counter) tensors; int64 wrap-around
        matches the masked python-int math bit-exactly (unit-tested).
        Advances the device draw counter; SamplerState.observe advances
        the host mirror so a composition-change rebuild stays in sync."""
        k = (self.seeds_dev * 6364136223846793005 +
             (self.draws_dev + 1) * 1442695040888963407) \
            & ((1 << 63) - 1)
        k = k * self.nongreedy_dev           # greedy rows draw key 0
        self.draws_dev += 1
        return k.unsqueeze(1)


_ARANGE_CACHE: dict = {}


def _arange(v: int, dev) -> torch.Tensor:
    key = (v, str(dev))
    t = _ARANGE_CACHE.get(key)
    if t is None:
        t = torch.arange(v, device=dev, dtype=torch.int64).unsqueeze(0)
        _ARANGE_CACHE[key] = t
    return t


def sample_batch_vectorized(logits: torch.Tensor,
                            states: list["SamplerState"],
                            plan: Optional[BatchPlan] = None) -> list[int]:
    """Batched sampling: penalties (per-row tensor ops, no syncs) ->
    temperature -> batched top_k/top_p/min_p filtering -> per-row Gumbel
    noise (seeded rows use their own generator, issued without a sync) ->
    ONE argmax + ONE host copy. The per-row `sample()` loop costs a GPU
    sync per ROW per step — at serving batch sizes that multiplies the
    decode step several-fold. Pass a persistent `plan` to reuse the
    per-row constant tensors across steps (engine does)."""
    dev = logits.device
    b, v = logits.shape
    if plan is None:
        plan = BatchPlan()
    plan.refresh(states, dev, v)
    # penalties on RAW logits (before temperature), matching sample().
    # ONE global sparse scatter for the whole batch: (row, token) pairs
    # gathered from every penalized row's seen-token dict — no per-row
    # GPU launches. Mutates `logits` rows in place (callers pass per-step
    # decode outputs that are fully rewritten before their next use).
    if plan.any_penalty:
        pi: list[int] = []
        pt: list[int] = []
        pc: list[float] = []
        prp: list[float] = []
        ppp: list[float] = []
        pfp: list[float] = []
        for i, s in enumerate(states):
            if not s.counts:
                continue
            o = s.opts
            rp = o.repeat_penalty if o.repeat_penalty else 1.0
            pp = o.presence_penalty or 0.0
            fp = o.frequency_penalty or 0.0
            for t_, c_ in s.counts.items():
                pi.append(i)
                pt.append(t_)
                pc.append(float(c_))
                prp.append(rp)
                ppp.append(pp)
                pfp.append(fp)
        if pi:
            ii = torch.tensor(pi, dtype=torch.long, device=dev)
            tt = torch.tensor(pt, dtype=torch.long, device=dev)
            cc = torch.tensor(pc, device=dev)
            rp_t = torch.tensor(prp, device=dev)
            pp_t = torch.tensor(ppp, device=dev)
            fp_t = torch.tensor(pfp, device=dev)
            vals = logits[ii, tt]
            vals = torch.where(vals > 0, vals / rp_t, vals * rp_t)
            vals = vals - pp_t - fp_t * cc
            logits[ii, tt] = vals
    if (logits.is_cuda and not plan.need_filter
            and os.environ.get("GRIDLLM_FUSED_SAMPLER", "1") != "0"):
        # DEFAULT ON (hardware-validated round 2: test green on MI355X,
        # serving +6% — gpurun_out/r02_lt_c192_fused.log): one HIP launch
        # replacing the ~15-op eager chain below; same RNG contract
        # (csrc k_sample_gumbel docstring). Greedy rows ride along with
        # key==0. GRIDLLM_FUSED_SAMPLER=0 restores the eager chain.
        from gridllm_amd.ops import hip_ops
        return hip_ops.sample_gumbel(
            logits.contiguous(),
            plan.keys(states, dev).view(-1).contiguous(),
            plan.temps.view(-1).contiguous()).tolist()
    x = logits / plan.temps
    if plan.need_filter:
        probs = torch.softmax(x, -1)
        sp, _ = torch.sort(probs, -1, descending=True)
        thr = sp.gather(1, plan.kidx)                       # top_k
        thr = torch.maximum(thr, sp[:, :1] * plan.mps)      # min_p
        cum = torch.cumsum(sp, -1)
        cnt = ((cum - sp) < plan.tps).sum(1, keepdim=True).clamp(min=1)
        thr = torch.maximum(thr, sp.gather(1, cnt - 1))     # top_p
        x = torch.where(probs >= thr, x, torch.full_like(x, float("-inf")))
    if plan.all_greedy:
        return x.argmax(-1).tolist()
    # batched Gumbel noise from a counter-based hash (splitmix64-style):
    # deterministic per (seed, draw index), zero python-per-row launches
    keys = plan.keys(states, dev)
    idx = _arange(v, dev)
    z = (idx ^ keys) * -7046029254386353131        # 0x9E3779B97F4A7C15
    z = z ^ (z >> 29)
    z = z * -4658895280553007687                   # 0xBF58476D1CE4E5B9
    z = z ^ (z >> 32)
    u = ((z >> 38) & 0xFFFFFF).to(torch.float32) / float(1 << 24)
    u = u.clamp(1e-6, 1.0 - 1e-6)
    noise = -torch.log(-torch.log(u))
    return (x + noise * plan.greedy_mask).argmax(-1).tolist()


def _apply_penalties(logits: torch.Tensor, states: list) -> None:
    """Sparse in-place penalty pass over penalized rows (host-gathered
    (row, token) pairs, one global scatter)."""
    dev = logits.device
    pi: list[int] = []
    pt: list[int] = []
    pc: list[float] = []
    prp: list[float] = []
    ppp: list[float] = []
    pfp: list[float] = []
    for i, s in enumerate(states):
        if not s.counts:
            continue
        o = s.opts
        rp = o.repeat_penalty if o.repeat_penalty else 1.0
        pp = o.presence_penalty or 0.0
        fp = o.frequency_penalty or 0.0
        for t_, c_ in s.counts.items():
            pi.append(i)
            pt.append(t_)
            pc.append(float(c_))
            prp.append(rp)
            ppp.append(pp)
            pfp.append(fp)
    if not pi:
        return
    ii = torch.tensor(pi, dtype=torch.long, device=dev)
    tt = torch.tensor(pt, dtype=torch.long, device=dev)
    cc = torch.tensor(pc, device=dev)
    rp_t = torch.tensor(prp, device=dev)
    pp_t = torch.tensor(ppp, device=dev)
    fp_t = torch.tensor(pfp, device=dev)
    vals = logits[ii, tt]
    vals = torch.where(vals > 0, vals / rp_t, vals * rp_t)
    vals = vals - pp_t - fp_t * cc
    logits[ii, tt] = vals


class SlotSampler:
    """Per-KV-SLOT device-resident sampling parameters.

    The BatchPlan cache keyed on batch composition still rebuilt (a
    per-row Python loop + H2D uploads) nearly EVERY step under serving
    churn — at c=192 each step retires and admits rows, so the
    composition never repeats (round-2 GPU profile: plan refresh 1.5 s of
    an 11 s decode-thread window). Here every per-row constant lives in a
    device array indexed by the row's KV slot:
      * admission writes the new rows' parameters once (one packed H2D +
        scatters);
      * each step gathers by the SAME slot-id tensor the decode step
        uses — zero host->device traffic, zero per-row Python;
      * the draw counter advances on-device; the splitmix key stream is
        bit-identical to the host-int math (unit-tested), so seeded
        determinism is unchanged.
    """

    def __init__(self, n_slots: int, device):
        self.device = device
        self._up = None
        self.temps = torch.full((n_slots,), 1.0, device=device)
        self.nongreedy = torch.zeros(n_slots, dtype=torch.int64,
                                     device=device)
        self.kidx = torch.zeros(n_slots, dtype=torch.int64, device=device)
        self.mps = torch.zeros(n_slots, device=device)
        self.tps = torch.ones(n_slots, device=device)
        self.seeds = torch.zeros(n_slots, dtype=torch.int64, device=device)
        self.draws = torch.zeros(n_slots, dtype=torch.int64, device=device)

    def admit(self, rows, vocab: int) -> None:
        """Register newly admitted rows (objects with .slot and .state)."""
        import numpy as np
        n = len(rows)
        host = np.zeros((5, n), dtype=np.float64)
        ints = np.zeros((3, n), dtype=np.int64)
        for i, r in enumerate(rows):
            s = r.state
            o = s.opts
            host[0, i] = max(o.temperature if o.temperature is not None
                             else 0.8, 1e-4)
            host[1, i] = 0.0 if s.greedy else 1.0
            host[2, i] = (int(o.top_k) - 1
                          if o.top_k and 0 < o.top_k < vocab else vocab - 1)
            host[3, i] = (o.min_p if o.min_p and 0 < o.min_p < 1 else 0.0)
            host[4, i] = (o.top_p if o.top_p and 0 < o.top_p < 1 else 1.0)
            ints[0, i] = r.slot
            ints[1, i] = s.noise_seed & ((1 << 63) - 1)
            ints[2, i] = s.draws
        if self._up is None:
            from gridllm_amd.utils.pinned import PinnedUploader
            self._up = PinnedUploader(self.device)
        t = self._up.upload(host)
        it = self._up.upload(ints)
        sl = it[0]
        self.temps[sl] = t[0].float()
        self.nongreedy[sl] = t[1].long()
        self.kidx[sl] = t[2].long()
        self.mps[sl] = t[3].float()
        self.tps[sl] = t[4].float()
        self.seeds[sl] = it[1]
        self.draws[sl] = it[2]

    def _keys(self, sl: torch.Tensor) -> torch.Tensor:
        k = (self.seeds[sl] * 6364136223846793005 +
             (self.draws[sl] + 1) * 1442695040888963407) & ((1 << 63) - 1)
        k = k * self.nongreedy[sl]
        self.draws[sl] += 1                  # unique slots: safe scatter
        return k

    def sample(self, logits: torch.Tensor, rows,
               slot_ids: torch.Tensor) -> list[int]:
        """Batched sampling; slot_ids: device int64 [B] aligned with rows
        (the decode step's own slot tensor)."""
        states = [r.state for r in rows]
        if any(s.counts for s in states):
            _apply_penalties(logits, states)
        return self.sample_ids(logits, rows, slot_ids).tolist()

    def sample_ids(self, logits: torch.Tensor, rows,
                   slot_ids: torch.Tensor) -> torch.Tensor:
        """Device-tensor sampling (no host sync): the pipelined decode
        path keeps the sampled ids ON DEVICE and feeds them into the next
        step. Penalty application (host-sparse) is the caller's concern;
        rows with penalties take the synchronous path instead."""
        states = [r.state for r in rows]
        if all(s.greedy for s in states):
            return logits.argmax(dim=-1)
        need_filter = any(s.need_filter for s in states)
        temps = self.temps[slot_ids].unsqueeze(1)
        if (logits.is_cuda and not need_filter
                and os.environ.get("GRIDLLM_FUSED_SAMPLER", "1") != "0"):
            from gridllm_amd.ops import hip_ops
            return hip_ops.sample_gumbel(
                logits.contiguous(), self._keys(slot_ids).contiguous(),
                temps.view(-1).contiguous())
        x = logits / temps
        v = logits.shape[1]
        if need_filter:
            probs = torch.softmax(x, -1)
            sp, _ = torch.sort(probs, -1, descending=True)
            thr = sp.gather(1, self.kidx[slot_ids].unsqueeze(1))
            thr = torch.maximum(thr, sp[:, :1]
                                * self.mps[slot_ids].unsqueeze(1))
            cum = torch.cumsum(sp, -1)
            cnt = ((cum - sp) < self.tps[slot_ids].unsqueeze(1)) \
                .sum(1, keepdim=True).clamp(min=1)
            thr = torch.maximum(thr, sp.gather(1, cnt - 1))
            x = torch.where(probs >= thr, x,
                            torch.full_like(x, float("-inf")))
        keys = self._keys(slot_ids).unsqueeze(1)
        idx = _arange(v, logits.device)
        z = (idx ^ keys) * -7046029254386353131
        z = z ^ (z >> 29)
        z = z * -4658895280553007687
        z = z ^ (z >> 32)
        u = ((z >> 38) & 0xFFFFFF).to(torch.float32) / float(1 << 24)
        u = u.clamp(1e-6, 1.0 - 1e-6)
        noise = -torch.log(-torch.log(u))
        gm = self.nongreedy[slot_ids].unsqueeze(1).float()
        return (x + noise * gm).argmax(-1)
