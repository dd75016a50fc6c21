#!/usr/bin/env python3
"""Flagship serving benchmark: llama3.2-3B bf16 batched streaming decode.

Measures the BASELINE.json headline metric — output tokens/sec for the whole
node — on synthetic prompts with random-init weights (no network for real
checkpoints). One rank per GPU (data-parallel request sharding, the
reference's only parallelism — SURVEY.md §2.4); each rank runs the full
engine decode path: hipGraph-replayed forward (HIP kernels + hipBLASLt
GEMMs) + batched sampling, exactly as the serving loop does.

Usage: python bench.py [--gpus N] [--steps K] [--warmup W] [--batch B]
Launched for N>1 via torch.distributed.run (reads RANK/WORLD_SIZE env).
"""

from __future__ import annotations

import argparse
import json
import os
import statistics
import time

import torch


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=32)
    ap.add_argument("--warmup", type=int, default=8)
    ap.add_argument("--batch", type=int, default=1024, help="sequences per GPU")
    ap.add_argument("--model", default="llama3.2:3b")
    ap.add_argument("--prompt-len", type=int, default=128)
    ap.add_argument("--max-seq", type=int, default=1024)
    ap.add_argument("--device", default="")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    n_gpus = world if world > 1 else args.gpus

    on_gpu = torch.cuda.is_available()
    device = args.device or (f"cuda:{local_rank}" if on_gpu else "cpu")
    if on_gpu:
        torch.cuda.set_device(device)

    dist = None
    if world > 1:
        import torch.distributed as tdist
        dist = tdist
        backend = "nccl" if on_gpu else "gloo"
        dist.init_process_group(backend=backend)

    from gridllm_amd.config import EngineConfig
    from gridllm_amd.engine import InferenceEngine
    from gridllm_amd.engine.sampler import SamplerState
    from gridllm_amd.types import SamplingOptions

    model = args.model if on_gpu else "tiny"
    batch = args.batch if on_gpu else 4
    # +1 slot so the TTFT measurement can allocate while the decode batch
    # holds `batch` slots
    cfg = EngineConfig(model=model, device=device,
                       dtype="bf16" if on_gpu else "fp32",
                       max_seq_len=args.max_seq, max_batch_size=batch + 1,
                       use_hipgraph=on_gpu)
    eng = InferenceEngine(cfg)

    # --- synthetic prompts: prefill `batch` slots with prompt_len random
    # tokens (batched prefill: chunks of 64 sequences per pass)
    torch.manual_seed(1000 + rank)
    slots = [eng.model.cache.alloc() for _ in range(batch)]
    vocab = eng.spec.vocab
    for i in range(0, batch, 64):
        chunk = slots[i:i + 64]
        toks = [torch.randint(0, vocab, (args.prompt_len,), device=eng.device)
                for _ in chunk]
        if len(chunk) > 1:
            eng.model.forward_prefill_batch(toks, chunk)
        else:
            eng.model.forward_prefill(toks[0], chunk[0])

    opts = SamplingOptions.from_dict({"temperature": 0.0})

    class _R:  # minimal row shims for the engine's batched sampler
        def __init__(self):
            self.state = SamplerState.create(opts, eng.device)
            self.constraint = None
    rows = [_R() for _ in range(batch)]
    tokens = [int(torch.randint(0, vocab, (1,)).item()) for _ in range(batch)]

    def step() -> list[int]:
        logits = eng._decode_step(tokens, slots)
        return eng._sample_batch(logits, rows)

    # --- warmup (also captures the hipGraph)
    for _ in range(args.warmup):
        tokens = step()

    def barrier_sync():
        if on_gpu:
            torch.cuda.synchronize()
        if dist is not None:
            dist.barrier()
        if on_gpu:
            torch.cuda.synchronize()

    barrier_sync()
    t0 = time.monotonic()
    for _ in range(args.steps):
        tokens = step()
    barrier_sync()
    t1 = time.monotonic()

    elapsed = t1 - t0
    if dist is not None:
        e = torch.tensor([elapsed], dtype=torch.float64)
        if on_gpu:
            e = e.to(device)
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = float(e.item())

    ms_per_step = elapsed * 1000.0 / args.steps
    total_tokens = args.steps * batch * n_gpus
    toks_per_sec = total_tokens / elapsed

    # --- p50 TTFT (prefill + first token), measured outside the timed region
    ttfts = []
    for _ in range(5):
        s = eng.model.cache.alloc()
        ids = torch.randint(0, vocab, (args.prompt_len,), device=eng.device)
        if on_gpu:
            torch.cuda.synchronize()
        tt = time.monotonic()
        logits = eng.model.forward_prefill(ids, s)
        _ = int(logits.argmax().item())
        ttfts.append((time.monotonic() - tt) * 1000.0)
        eng.model.cache.free(s)
    ttft_p50 = statistics.median(ttfts)

    if rank == 0:
        out = {
            "metric": "output tokens/sec (node), llama3.2 decode",
            "value": round(toks_per_sec, 1),
            "unit": "tokens/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": cfg.dtype,
            "data": "synthetic",
            "config": {
                "model": model,
                "global_batch": batch * n_gpus,
                "seq_len": args.prompt_len,
                "max_seq": args.max_seq,
                "parallelism": f"dp{n_gpus}",
                "ttft_p50_ms": round(ttft_p50, 2),
            },
        }
        print(json.dumps(out))

    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
