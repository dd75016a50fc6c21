"""Llama-family transformer, MI355X-first.

Design (no reference analog — the reference delegated all compute to Ollama,
client/src/services/OllamaService.ts):

* fused weight layout: one QKV GEMM, one gate+up GEMM per layer. Decode
  GEMMs dispatch through a per-shape autotuner (ops.autotune: hipBLASLt vs
  the in-house skinny / glds-pipelined projection kernels); the gate+up
  projection runs with a fused-SwiGLU epilogue on an interleaved weight
  layout. Everything between GEMMs is a hand-written CDNA4 kernel
  (gridllm_amd.ops): fused residual-add RMSNorm, fused RoPE+KV-write, MFMA
  flash prefill, GQA decode attention (4-wave + one-wave large-batch
  variants, split-S for long sequences).
* KV cache: contiguous [slots, Hkv, max_seq, D] bf16 per layer, sized for
  288 GB HBM3E; slot indirection (slot_map) keeps decode shapes static so
  the whole decode step is hipGraph-capturable.
* tensor parallel: head-sharded QKV + row-parallel O / down with one RCCL
  all-reduce after each (2/layer), KV sharded by head across ranks.
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn.functional as F  # noqa: F401

from gridllm_amd.ops.autotune import linear as _linear
from gridllm_amd.ops.autotune import linear_gateup as _linear_gateup

from gridllm_amd import ops
from gridllm_amd.models.registry import ModelSpec
from gridllm_amd.parallel.tp import TPContext

_DTYPES = {"bf16": torch.bfloat16, "fp16": torch.float16, "fp32": torch.float32}


@dataclass
class LayerWeights:
    wqkv: torch.Tensor       # [(Hq_l + 2*Hkv_l) * D, H]
    wo: torch.Tensor         # [H, Hq_l * D]
    wgateup: torch.Tensor    # [2 * F_l, H]
    wdown: torch.Tensor      # [H, F_l]
    attn_norm: torch.Tensor  # [H]
    mlp_norm: torch.Tensor   # [H]
    bqkv: object = None      # [(Hq_l + 2*Hkv_l) * D] Qwen2-family q/k/v bias


class KVCache:
    """Slot-based KV cache: [slots, Hkv_local, max_seq, D] bf16 per layer."""

    def __init__(self, spec: ModelSpec, slots: int, max_seq: int,
                 hkv_local: int, device: torch.device, dtype: torch.dtype):
        self.slots = slots
        self.max_seq = max_seq
        shape = (slots, hkv_local, max_seq, spec.head_dim)
        self.k = [torch.zeros(shape, dtype=dtype, device=device)
                  for _ in range(spec.layers)]
        self.v = [torch.zeros(shape, dtype=dtype, device=device)
                  for _ in range(spec.layers)]
        self.seq_lens = torch.zeros(slots, dtype=torch.int32, device=device)
        # context shift: per-slot rope-position offset — after a shift the
        # kept K entries stay roped at their ORIGINAL absolute positions,
        # so new tokens must rope at storage_pos + rope_off (read inside
        # the hipGraph; plain tensor -> capture-safe)
        self.rope_off = torch.zeros(slots, dtype=torch.int32, device=device)
        self._free = list(range(slots))

    def reserve(self, slot: int) -> None:
        """Remove a slot from the allocatable pool (hipGraph scratch slot)."""
        if slot in self._free:
            self._free.remove(slot)

    def alloc(self) -> int:
        """HOST-ONLY bookkeeping: seq_lens/rope_off for the slot are
        (re)initialized by the next prefill's batched scatter — a per-slot
        scalar device write here measured ~0.5 ms of stream-sync stall per
        call under serving load (round-2 GPU profile)."""
        if not self._free:
            raise RuntimeError("KV cache: no free slots")
        return self._free.pop()

    def free(self, slot: int) -> None:
        self._free.append(slot)

    @property
    def n_free(self) -> int:
        return len(self._free)

    def bytes(self) -> int:
        return sum(t.numel() * t.element_size() for t in self.k) * 2


class LlamaModel:
    def __init__(self, spec: ModelSpec, device: str = "cpu",
                 dtype: str = "bf16", tp: Optional[TPContext] = None,
                 seed: int = 0, kv_slots: int = 64, max_seq: int = 0):
        self.spec = spec
        self.device = torch.device(device)
        self.dtype = _DTYPES[dtype]
        self.tp = tp or TPContext()
        self.max_seq = max_seq or spec.max_seq_len

        assert spec.heads % self.tp.size == 0
        assert spec.kv_heads % self.tp.size == 0
        assert spec.intermediate % self.tp.size == 0
        self.hq_l = spec.heads // self.tp.size
        self.hkv_l = spec.kv_heads // self.tp.size
        self.f_l = spec.intermediate // self.tp.size
        self.scale = 1.0 / math.sqrt(spec.head_dim)

        self.gateup_il = False
        self._init_weights(seed)
        # tables cover 4x the window: context shift keeps original absolute
        # rope positions, so a long generation's positions exceed max_seq
        self.rope_len = 4 * self.max_seq
        self.cos, self.sin = ops.rope_tables(
            spec.head_dim, self.rope_len, spec.rope_theta, device=self.device)
        self.cache = KVCache(spec, kv_slots, self.max_seq, self.hkv_l,
                             self.device, self.dtype if self.dtype != torch.float32
                             else torch.float32)
        from gridllm_amd.utils.pinned import PinnedUploader
        self.up = PinnedUploader(self.device)   # stall-free H2D staging

    # -- weights -----------------------------------------------------------

    def _randn(self, gen: torch.Generator, *shape: int,
               std: float = 0.02) -> torch.Tensor:
        t = torch.randn(*shape, generator=gen, device=self.device,
                        dtype=torch.float32) * std
        return t.to(self.dtype)

    def _init_weights(self, seed: int) -> None:
        """Random init; TP ranks generate the FULL tensor with the same seed
        and slice their shard, so any tp degree realizes the same model."""
        s, tp = self.spec, self.tp
        d = s.head_dim
        gen = torch.Generator(device=self.device)
        gen.manual_seed(seed or 1234)

        self.embed = self._randn(gen, s.vocab, s.hidden)
        self.final_norm = torch.ones(s.hidden, dtype=self.dtype, device=self.device)
        if s.tie_embeddings:
            self.lm_head = self.embed
        else:
            self.lm_head = self._randn(gen, s.vocab, s.hidden)

        q0, _ = tp.shard(s.heads)
        k0, _ = tp.shard(s.kv_heads)
        f0, _ = tp.shard(s.intermediate)
        self.layers: list[LayerWeights] = []
        for _ in range(s.layers):
            wq = self._randn(gen, s.heads * d, s.hidden)
            wk = self._randn(gen, s.kv_heads * d, s.hidden)
            wv = self._randn(gen, s.kv_heads * d, s.hidden)
            wo_full = self._randn(gen, s.hidden, s.heads * d)
            wg = self._randn(gen, s.intermediate, s.hidden)
            wu = self._randn(gen, s.intermediate, s.hidden)
            wd_full = self._randn(gen, s.hidden, s.intermediate)
            wqkv = torch.cat([
                wq[q0 * d:(q0 + self.hq_l) * d],
                wk[k0 * d:(k0 + self.hkv_l) * d],
                wv[k0 * d:(k0 + self.hkv_l) * d],
            ]).contiguous()
            wo = wo_full[:, q0 * d:(q0 + self.hq_l) * d].contiguous()
            wgateup = torch.cat([wg[f0:f0 + self.f_l],
                                 wu[f0:f0 + self.f_l]]).contiguous()
            wdown = wd_full[:, f0:f0 + self.f_l].contiguous()
            bqkv = None
            if s.qkv_bias:
                bq = self._randn(gen, s.heads * d).flatten()
                bk = self._randn(gen, s.kv_heads * d).flatten()
                bv = self._randn(gen, s.kv_heads * d).flatten()
                bqkv = torch.cat([
                    bq[q0 * d:(q0 + self.hq_l) * d],
                    bk[k0 * d:(k0 + self.hkv_l) * d],
                    bv[k0 * d:(k0 + self.hkv_l) * d]]).contiguous()
            self.layers.append(LayerWeights(
                wqkv=wqkv, wo=wo, wgateup=wgateup, wdown=wdown, bqkv=bqkv,
                attn_norm=torch.ones(s.hidden, dtype=self.dtype, device=self.device),
                mlp_norm=torch.ones(s.hidden, dtype=self.dtype, device=self.device)))

    def interleave_gateup(self) -> None:
        """Reorder each wgateup's rows into 128-row groups [64 gate |
        64 up] so the fused-SwiGLU projection kernel (ops.gemm_proj with
        fuse_swiglu) finds matching gate/up column pairs in every BN=128
        block. GPU-only layout change; forward dispatches accordingly."""
        if self.gateup_il or self.f_l % 64 != 0:
            return
        f = self.f_l
        g = torch.arange(f)
        perm = torch.empty(2 * f, dtype=torch.int64)
        perm[(g // 64) * 128 + g % 64] = g          # gate rows
        perm[(g // 64) * 128 + g % 64 + 64] = f + g  # up rows
        perm = perm.to(self.device)
        for lw in self.layers:
            lw.wgateup = lw.wgateup[perm].contiguous()
        self.gateup_il = True

    def _mlp_act(self, normed: torch.Tensor, lw: LayerWeights) -> torch.Tensor:
        if self.gateup_il:
            return _linear_gateup(normed, lw.wgateup)
        return ops.swiglu(_linear(normed, lw.wgateup))

    def load_safetensors(self, path: str) -> int:
        """Load HF-layout llama weights (model.safetensors or sharded
        model-*-of-*.safetensors in `path`) into the fused TP-sharded
        layout. Each rank reads only its shard slice. Returns the number
        of tensors consumed. (the engine-side analog of
        checkpoint/resume — the reference had none, weights lived inside
        Ollama.)"""
        import glob as _glob
        import os
        from safetensors import safe_open

        self.gateup_il = False          # loads land in the plain layout
        files = sorted(_glob.glob(os.path.join(path, "*.safetensors")))
        if not files:
            raise FileNotFoundError(f"no .safetensors under {path!r}")
        tensors: dict[str, torch.Tensor] = {}
        want = self._hf_key_set()
        for f in files:
            with safe_open(f, framework="pt") as sf:
                for k in sf.keys():
                    if k in want:
                        tensors[k] = sf.get_tensor(k)
        missing = want - set(tensors)
        if missing:
            raise KeyError(f"missing weights: {sorted(missing)[:5]} "
                           f"(+{max(0, len(missing) - 5)} more)")
        self._apply_hf(tensors)
        return len(tensors)

    def _hf_key_set(self) -> set:
        s = self.spec
        keys = {"model.embed_tokens.weight", "model.norm.weight"}
        if not s.tie_embeddings:
            keys.add("lm_head.weight")
        for i in range(s.layers):
            p = f"model.layers.{i}."
            keys |= {p + "self_attn.q_proj.weight",
                     p + "self_attn.k_proj.weight",
                     p + "self_attn.v_proj.weight",
                     p + "self_attn.o_proj.weight",
                     p + "mlp.gate_proj.weight", p + "mlp.up_proj.weight",
                     p + "mlp.down_proj.weight",
                     p + "input_layernorm.weight",
                     p + "post_attention_layernorm.weight"}
            if s.qkv_bias:
                keys |= {p + "self_attn.q_proj.bias",
                         p + "self_attn.k_proj.bias",
                         p + "self_attn.v_proj.bias"}
        return keys

    def _apply_hf(self, t: dict) -> None:
        s, tp, d = self.spec, self.tp, self.spec.head_dim
        dt, dev = self.dtype, self.device

        def cv(x: torch.Tensor) -> torch.Tensor:
            return x.to(device=dev, dtype=dt).contiguous()

        self.embed = cv(t["model.embed_tokens.weight"])
        self.final_norm = cv(t["model.norm.weight"])
        self.lm_head = self.embed if s.tie_embeddings \
            else cv(t["lm_head.weight"])
        q0, _ = tp.shard(s.heads)
        k0, _ = tp.shard(s.kv_heads)
        f0, _ = tp.shard(s.intermediate)
        for i, lw in enumerate(self.layers):
            p = f"model.layers.{i}."
            wq = t[p + "self_attn.q_proj.weight"]
            wk = t[p + "self_attn.k_proj.weight"]
            wv = t[p + "self_attn.v_proj.weight"]
            lw.wqkv = cv(torch.cat([
                wq[q0 * d:(q0 + self.hq_l) * d],
                wk[k0 * d:(k0 + self.hkv_l) * d],
                wv[k0 * d:(k0 + self.hkv_l) * d]]))
            lw.wo = cv(t[p + "self_attn.o_proj.weight"]
                       [:, q0 * d:(q0 + self.hq_l) * d])
            lw.wgateup = cv(torch.cat([
                t[p + "mlp.gate_proj.weight"][f0:f0 + self.f_l],
                t[p + "mlp.up_proj.weight"][f0:f0 + self.f_l]]))
            lw.wdown = cv(t[p + "mlp.down_proj.weight"][:, f0:f0 + self.f_l])
            lw.attn_norm = cv(t[p + "input_layernorm.weight"])
            lw.mlp_norm = cv(t[p + "post_attention_layernorm.weight"])
            if s.qkv_bias:
                lw.bqkv = cv(torch.cat([
                    t[p + "self_attn.q_proj.bias"]
                    [q0 * d:(q0 + self.hq_l) * d],
                    t[p + "self_attn.k_proj.bias"]
                    [k0 * d:(k0 + self.hkv_l) * d],
                    t[p + "self_attn.v_proj.bias"]
                    [k0 * d:(k0 + self.hkv_l) * d]]))

    def param_bytes(self) -> int:
        total = self.embed.numel() + self.final_norm.numel()
        if self.lm_head is not self.embed:
            total += self.lm_head.numel()
        for lw in self.layers:
            for t in (lw.wqkv, lw.wo, lw.wgateup, lw.wdown, lw.attn_norm,
                      lw.mlp_norm):
                total += t.numel()
            if lw.bqkv is not None:
                total += lw.bqkv.numel()
        return total * self.embed.element_size()

    def shift_slot(self, slot: int) -> int:
        """Context shift (the reference delegated this to Ollama/llama.cpp's
        num_ctx handling): drop the oldest half of the slot's KV, keep the
        recent half AT ITS ORIGINAL ROPE POSITIONS (rope_off grows by the
        drop count, so relative distances are exact). Returns the new
        storage length. Eager, per-row, rare — no kernel needed."""
        keep = self.max_seq // 2
        drop = self.max_seq - keep
        for li in range(self.spec.layers):
            k = self.cache.k[li]
            v = self.cache.v[li]
            k[slot, :, :keep] = k[slot, :, drop:].clone()
            v[slot, :, :keep] = v[slot, :, drop:].clone()
        self.cache.seq_lens[slot] = keep
        self.cache.rope_off[slot] += drop
        return keep

    # -- forward -----------------------------------------------------------

    @torch.no_grad()
    def forward_prefill(self, tokens: torch.Tensor, slot: int,
                        start_pos: int = 0,
                        return_hidden: bool = False) -> torch.Tensor:
        """Prefill one sequence. tokens: [T] int64 on device.
        Returns logits [vocab] (f32) of the last position; with
        return_hidden, returns (logits, mean-pooled final hidden [H] f32)."""
        s = self.spec
        t = tokens.shape[0]
        pos = torch.arange(start_pos, start_pos + t, dtype=torch.int32,
                           device=self.device)
        slot_t = torch.full((t,), slot * self.cache.max_seq, dtype=torch.int32,
                            device=self.device) + pos
        x = self.embed[tokens]                              # [T, H]
        resid = x
        normed, _ = ops.rmsnorm(x, self.layers[0].attn_norm, s.norm_eps)
        for li, lw in enumerate(self.layers):
            qkv = _linear(normed, lw.wqkv)
            if lw.bqkv is not None:
                qkv = qkv + lw.bqkv
            q = ops.rope_kv(qkv, self.cache.k[li], self.cache.v[li],
                            self.cos, self.sin, pos, slot_t,
                            self.hq_l, self.hkv_l, self.spec.head_dim)
            attn = ops.attn_prefill_cache(
                q, self.cache.k[li], self.cache.v[li], slot, start_pos + t,
                start_pos, self.scale)                           # [T,Hq_l,D]
            attn_out = _linear(attn.reshape(t, -1), lw.wo)
            self.tp.all_reduce(attn_out)
            normed, resid = ops.rmsnorm(attn_out, lw.mlp_norm, s.norm_eps,
                                        residual=resid)
            act = self._mlp_act(normed, lw)
            mlp_out = _linear(act, lw.wdown)
            self.tp.all_reduce(mlp_out)
            next_norm = (self.layers[li + 1].attn_norm
                         if li + 1 < len(self.layers) else self.final_norm)
            normed, resid = ops.rmsnorm(mlp_out, next_norm, s.norm_eps,
                                        residual=resid)
        self.cache.seq_lens[slot] = start_pos + t
        if start_pos == 0:
            # fresh slot: reset the shift offset (alloc is host-only)
            self.cache.rope_off[slot] = 0
        logits = _linear(normed[-1:], self.lm_head).float()
        if return_hidden:
            return logits[0], normed.float().mean(dim=0)
        return logits[0]

    @torch.no_grad()
    def forward_prefill_batch(self, token_lists: list[torch.Tensor],
                              slots: list[int],
                              return_hidden: bool = False) -> torch.Tensor:
        """Prefill SEVERAL fresh sequences in one pass: tokens are
        concatenated so every GEMM runs at M = sum(T_i) instead of k tiny
        launches (the admission-burst TTFT cost under serving load);
        attention stays per-sequence against each slot. Returns last-token
        logits [k, vocab] f32; with return_hidden, (logits, mean-pooled
        final hidden [k, H] f32) — the embedding path."""
        s = self.spec
        import numpy as np
        lens = [int(t.shape[0]) for t in token_lists]
        bounds = [0]
        for ln in lens:
            bounds.append(bounds[-1] + ln)
        tokens = torch.cat(token_lists)
        # position/slot indices built host-side: TWO H2D copies instead of
        # ~3k tiny device aranges per admission batch
        pos_np = np.concatenate(
            [np.arange(ln, dtype=np.int32) for ln in lens])
        slot_np = np.concatenate(
            [sl * self.cache.max_seq + np.arange(ln, dtype=np.int32)
             for ln, sl in zip(lens, slots)])
        idx = self.up.upload(np.stack([pos_np, slot_np]))
        pos, slot_t = idx[0], idx[1]
        # varlen batched attention (one meta-driven kernel launch per layer
        # instead of a per-sequence Python loop — the round-2 profile's
        # 36 ms-per-admission hot spot); meta rows: {q_off, len, slot, q0}
        meta = None
        if (tokens.is_cuda and self.dtype == torch.bfloat16
                and s.head_dim in (64, 128)):
            rows = [(bounds[i], ln, sl, q0)
                    for i, (ln, sl) in enumerate(zip(lens, slots))
                    for q0 in range(0, ln, 128)]
            meta = self.up.upload(np.asarray(rows, dtype=np.int32))
        x = self.embed[tokens]
        resid = x
        normed, _ = ops.rmsnorm(x, self.layers[0].attn_norm, s.norm_eps)
        for li, lw in enumerate(self.layers):
            qkv = _linear(normed, lw.wqkv)
            if lw.bqkv is not None:
                qkv = qkv + lw.bqkv
            q = ops.rope_kv(qkv, self.cache.k[li], self.cache.v[li],
                            self.cos, self.sin, pos, slot_t,
                            self.hq_l, self.hkv_l, self.spec.head_dim)
            if meta is not None:
                attn = ops.attn_prefill_cache_batch(
                    q, self.cache.k[li], self.cache.v[li], meta, self.scale)
            else:
                attn = torch.empty_like(q)
                for i, sl in enumerate(slots):
                    attn[bounds[i]:bounds[i + 1]] = ops.attn_prefill_cache(
                        q[bounds[i]:bounds[i + 1]], self.cache.k[li],
                        self.cache.v[li], sl, lens[i], 0, self.scale)
            attn_out = _linear(attn.reshape(bounds[-1], -1), lw.wo)
            self.tp.all_reduce(attn_out)
            normed, resid = ops.rmsnorm(attn_out, lw.mlp_norm, s.norm_eps,
                                        residual=resid)
            act = self._mlp_act(normed, lw)
            mlp_out = _linear(act, lw.wdown)
            self.tp.all_reduce(mlp_out)
            next_norm = (self.layers[li + 1].attn_norm
                         if li + 1 < len(self.layers) else self.final_norm)
            normed, resid = ops.rmsnorm(mlp_out, next_norm, s.norm_eps,
                                        residual=resid)
        # ONE scatter instead of k per-sequence device writes (each is a
        # serialized H2D launch — 15 ms of host time per admission batch
        # in the round-2 serving profile)
        import numpy as _np
        host = _np.asarray([[sl for sl in slots], lens,
                            [b - 1 for b in bounds[1:]]], dtype=_np.int64)
        dev3 = self.up.upload(host)
        self.cache.seq_lens[dev3[0]] = dev3[1].to(torch.int32)
        # fresh slots start unshifted (alloc no longer touches the device)
        self.cache.rope_off.index_fill_(0, dev3[0], 0)
        lasts = dev3[2]
        logits = _linear(normed[lasts], self.lm_head).float()
        if return_hidden:
            hidden = torch.stack([
                normed[bounds[i]:bounds[i + 1]].float().mean(dim=0)
                for i in range(len(token_lists))])
            return logits, hidden
        return logits

    @torch.no_grad()
    def forward_decode(self, tokens: torch.Tensor,
                       slot_map: torch.Tensor) -> torch.Tensor:
        """One decode step for a batch. tokens: [B] int64; slot_map: [B] int32.
        Appends to the cache, advances seq_lens, returns logits [B, vocab] f32.
        Fully tensor-driven -> hipGraph-capturable."""
        s = self.spec
        b = tokens.shape[0]
        pos = self.cache.seq_lens[slot_map.long()]          # [B] int32
        slot_t = slot_map * self.cache.max_seq + pos
        attn_seq = pos + 1
        # rope at the ABSOLUTE position (shifted slots have rope_off > 0)
        pos_rope = pos + self.cache.rope_off[slot_map.long()]
        x = self.embed[tokens]
        resid = x
        normed, _ = ops.rmsnorm(x, self.layers[0].attn_norm, s.norm_eps)
        for li, lw in enumerate(self.layers):
            qkv = _linear(normed, lw.wqkv)
            if lw.bqkv is not None:
                qkv = qkv + lw.bqkv
            q = ops.rope_kv(qkv, self.cache.k[li], self.cache.v[li],
                            self.cos, self.sin, pos_rope, slot_t,
                            self.hq_l, self.hkv_l, self.spec.head_dim)
            attn = ops.attn_decode(q, self.cache.k[li], self.cache.v[li],
                                   attn_seq, self.scale, slot_map=slot_map)
            attn_out = _linear(attn.reshape(b, -1), lw.wo)
            self.tp.all_reduce(attn_out)
            normed, resid = ops.rmsnorm(attn_out, lw.mlp_norm, s.norm_eps,
                                        residual=resid)
            act = self._mlp_act(normed, lw)
            mlp_out = _linear(act, lw.wdown)
            self.tp.all_reduce(mlp_out)
            next_norm = (self.layers[li + 1].attn_norm
                         if li + 1 < len(self.layers) else self.final_norm)
            normed, resid = ops.rmsnorm(mlp_out, next_norm, s.norm_eps,
                                        residual=resid)
        self.cache.seq_lens.index_add_(
            0, slot_map.long(),
            torch.ones(b, dtype=torch.int32, device=self.device))
        logits = _linear(normed, self.lm_head).float()
        return logits
