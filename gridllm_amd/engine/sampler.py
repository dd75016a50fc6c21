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


def sample_batch_vectorized(logits: torch.Tensor,
                            states: list["SamplerState"]) -> list[int]:
    """Batched sampling: penalties (per-row tensor ops, no syncs) ->
    temperature -> batched top_k/top_p/min_p filtering -> per-row Gumbel
    noise (seeded rows use their own generator, issued without a sync) ->
    ONE argmax + ONE host copy. The per-row `sample()` loop costs a GPU
    sync per ROW per step — at serving batch sizes that multiplies the
    decode step several-fold."""
    dev = logits.device
    b, v = logits.shape
    # penalties on RAW logits (before temperature), matching sample().
    # ONE global sparse scatter for the whole batch: (row, token) pairs
    # gathered from every penalized row's seen-token dict — no per-row
    # GPU launches. Mutates `logits` rows in place (callers pass per-step
    # decode outputs that are fully rewritten before their next use).
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
    temps = torch.tensor(
        [max(s.opts.temperature if s.opts.temperature is not None else 0.8,
             1e-4) for s in states], device=dev).unsqueeze(1)
    x = logits / temps
    need_filter = any((s.opts.top_k and 0 < s.opts.top_k < v)
                      or (s.opts.top_p and 0.0 < s.opts.top_p < 1.0)
                      or (s.opts.min_p and 0.0 < s.opts.min_p < 1.0)
                      for s in states)
    if need_filter:
        probs = torch.softmax(x, -1)
        sp, _ = torch.sort(probs, -1, descending=True)
        thr = torch.zeros(b, 1, device=dev)
        kidx = torch.tensor(
            [int(s.opts.top_k) - 1 if s.opts.top_k and 0 < s.opts.top_k < v
             else v - 1 for s in states], device=dev).unsqueeze(1)
        thr = torch.maximum(thr, sp.gather(1, kidx))
        mps = torch.tensor(
            [s.opts.min_p if s.opts.min_p and 0 < s.opts.min_p < 1 else 0.0
             for s in states], device=dev).unsqueeze(1)
        thr = torch.maximum(thr, sp[:, :1] * mps)
        tps = torch.tensor(
            [s.opts.top_p if s.opts.top_p and 0 < s.opts.top_p < 1 else 1.0
             for s in states], device=dev).unsqueeze(1)
        cum = torch.cumsum(sp, -1)
        cnt = ((cum - sp) < tps).sum(1, keepdim=True).clamp(min=1)
        thr = torch.maximum(thr, sp.gather(1, cnt - 1))
        x = torch.where(probs >= thr, x, torch.full_like(x, float("-inf")))
    # batched Gumbel noise from a counter-based hash (splitmix64-style):
    # deterministic per (seed, draw index), zero python-per-row launches
    keys = torch.tensor(
        [0 if s.greedy else
         ((s.noise_seed * 6364136223846793005 +
           (s.draws + 1) * 1442695040888963407) & (2**63 - 1))
         for s in states], device=dev, dtype=torch.int64).unsqueeze(1)
    idx = torch.arange(v, device=dev, dtype=torch.int64).unsqueeze(0)
    z = (idx ^ keys) * -7046029254386353131        # 0x9E3779B97F4A7C15
    z = z ^ (z >> 29)
    z = z * -4658895280553007687                   # 0xBF58476D1CE4E5B9
    z = z ^ (z >> 32)
    u = ((z >> 38) & 0xFFFFFF).to(torch.float32) / float(1 << 24)
    u = u.clamp(1e-6, 1.0 - 1e-6)
    noise = -torch.log(-torch.log(u))
    greedy_mask = torch.tensor([1.0 if not s.greedy else 0.0
                                for s in states], device=dev).unsqueeze(1)
    return (x + noise * greedy_mask).argmax(-1).tolist()
