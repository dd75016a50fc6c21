"""Ollama-compatible API (reference: server/src/routes/ollama.ts).

Endpoints: /api/generate, /api/chat, /api/tags, /api/embed, /api/embeddings
(mounted at both /ollama/api/* — the reference's mount point, index.ts:67 —
and /api/* for drop-in Ollama SDK compatibility).

Wire behavior preserved from the reference:
  * empty prompt + keep_alive==0 -> synthetic unload response; empty prompt
    alone -> synthetic load response (ollama.ts:177-214)
  * streaming = newline-delimited JSON over chunked transfer (:248-303)
  * /api/tags aggregates worker model lists + gridllm_metadata
    (:507-571)
  * 404 when no worker advertises the model (:120-129)
Difference: /api/chat performs REAL chat templating instead of the
reference's "role: content" flattening (ollama.ts:366-370) — response
shapes are identical.
"""

from __future__ import annotations

import asyncio
import json
import time
from datetime import datetime, timezone
from typing import Optional

from fastapi import APIRouter, HTTPException, Request
from fastapi.responses import JSONResponse, StreamingResponse

from gridllm_amd.server import tools as tools_mod
from gridllm_amd.server.registry import WorkerRegistry
from gridllm_amd.server.scheduler import JobScheduler
from gridllm_amd.types import InferenceRequest

_SAMPLING_KEYS = {
    "temperature", "top_p", "top_k", "min_p", "seed", "stop", "num_predict",
    "num_ctx", "repeat_penalty", "presence_penalty", "frequency_penalty",
}


def _now_iso() -> str:
    return datetime.now(timezone.utc).isoformat().replace("+00:00", "Z")


def _clean_options(options: Optional[dict]) -> dict:
    if not isinstance(options, dict):
        return {}
    from gridllm_amd.types import SamplingOptions
    err = SamplingOptions.validate(options)
    if err:                        # Ollama 400s wrongly-typed options at
        raise HTTPException(       # unmarshal; without this the junk
            status_code=400, detail=err)   # surfaced as an engine error
    out = {k: v for k, v in options.items() if k in _SAMPLING_KEYS}
    stop = out.get("stop")
    if isinstance(stop, str):
        out["stop"] = [stop]
    return out


def _timings(res: dict) -> dict:
    t = res.get("timings") or {}
    return {
        "total_duration": int(t.get("total_duration", 0)),
        "load_duration": int(t.get("load_duration", 0)),
        "prompt_eval_count": int(t.get("prompt_eval_count", 0)),
        "prompt_eval_duration": int(t.get("prompt_eval_duration", 0)),
        "eval_count": int(t.get("eval_count", 0)),
        "eval_duration": int(t.get("eval_duration", 0)),
    }


def _validate_model(registry: WorkerRegistry, model: str) -> None:
    if not model or not isinstance(model, str):
        raise HTTPException(status_code=400, detail="model is required")
    if not registry.get_workers_by_model(model):
        raise HTTPException(
            status_code=404,
            detail=f"model '{model}' not found — no worker advertises it")


def _check_prompt(body: dict) -> str:
    prompt = body.get("prompt", "")
    if not isinstance(prompt, str):
        raise HTTPException(status_code=400, detail="prompt must be a string")
    if len(prompt) > 100_000:  # reference cap (ollama.ts:19)
        raise HTTPException(status_code=400, detail="prompt too long (max 100kB)")
    return prompt


def _stream_ndjson(scheduler: JobScheduler, req: InferenceRequest,
                   to_chunk, to_final):
    """Shared NDJSON streaming plumbing for generate/chat."""

    async def gen():
        from gridllm_amd.server import metrics
        q: asyncio.Queue = asyncio.Queue()
        t0 = time.monotonic()
        first = [True]

        def on_chunk(c: dict) -> None:
            if first[0]:
                first[0] = False
                metrics.observe_ttft(time.monotonic() - t0)
            q.put_nowait(("chunk", c))

        async def run():
            try:
                res = await scheduler.submit_streaming(req, on_chunk)
                q.put_nowait(("final", res))
            except Exception as e:  # noqa: BLE001
                q.put_nowait(("error", str(e)))

        task = asyncio.get_running_loop().create_task(run())
        try:
            while True:
                kind, item = await q.get()
                if kind == "chunk":
                    if item.get("response") or item.get("thinking"):
                        yield json.dumps(to_chunk(item)) + "\n"
                    if item.get("done"):
                        continue  # final payload carries the done record
                elif kind == "final":
                    if item.get("error"):
                        yield json.dumps({"error": item["error"],
                                          "done": True}) + "\n"
                    else:
                        metrics.add_tokens(item.get("eval_count"))
                        yield json.dumps(to_final(item)) + "\n"
                    break
                else:
                    yield json.dumps({"error": item, "done": True}) + "\n"
                    break
        finally:
            task.cancel()
            if not task.done():
                # client disconnected mid-stream: abort the job on the
                # worker too, or it keeps generating to completion (the
                # SSE path got this in round 1 — advisor finding — but
                # the NDJSON path did not)
                try:
                    await scheduler.cancel_job(req.id)
                except Exception:  # noqa: BLE001
                    pass

    return StreamingResponse(gen(), media_type="application/x-ndjson")


def build_router(scheduler: JobScheduler, registry: WorkerRegistry) -> APIRouter:
    r = APIRouter()

    # ------------------------------------------------------------- generate

    @r.post("/api/generate")
    async def generate(request: Request):
        body = await request.json()
        model = body.get("model", "")
        prompt = _check_prompt(body)
        stream = bool(body.get("stream", True))
        keep_alive = body.get("keep_alive")

        # synthetic load/unload (reference ollama.ts:177-214)
        if prompt == "":
            _validate_model(registry, model)
            resp = {"model": model, "created_at": _now_iso(), "response": "",
                    "done": True}
            resp["done_reason"] = "unload" if keep_alive in (0, "0") else "load"
            return JSONResponse(resp)

        _validate_model(registry, model)
        opts = _clean_options(body.get("options"))
        fmt = body.get("format")
        if fmt == "json" or isinstance(fmt, dict):
            opts["format"] = fmt       # engine enforces via a byte-level
                                       # JSON automaton (engine/constraints)
        req = InferenceRequest(
            model=model, prompt=prompt, request_type="generate",
            stream=stream, priority=str(body.get("priority", "medium")),
            timeout_ms=300_000, options=opts,
            metadata={k: body[k] for k in
                      ("suffix", "images", "think", "format", "system",
                       "template", "raw", "keep_alive", "context")
                      if k in body},
        )
        if body.get("system") and not body.get("raw"):
            # raw=true bypasses all prompt shaping (Ollama semantics)
            req.prompt = f"{body['system']}\n\n{req.prompt}"

        def chunk_json(c):
            out = {"model": model, "created_at": _now_iso(),
                   "response": c.get("response", ""), "done": False}
            if c.get("thinking"):
                out["thinking"] = c["thinking"]
            return out

        def final_json(res):
            return {"model": model, "created_at": _now_iso(), "response": "",
                    "done": True,
                    "done_reason": res.get("done_reason", "stop"),
                    "context": res.get("context") or [], **_timings(res)}

        if stream:
            return _stream_ndjson(scheduler, req, chunk_json, final_json)
        res = await scheduler.submit_and_wait(req)
        if res.get("error"):
            raise HTTPException(status_code=500, detail=res["error"])
        out = {
            "model": model, "created_at": _now_iso(),
            "response": res.get("response", ""), "done": True,
            "done_reason": res.get("done_reason", "stop"),
            "context": res.get("context") or [],
            **_timings(res)}
        if res.get("thinking") is not None:
            out["thinking"] = res["thinking"]
        return JSONResponse(out)

    # ----------------------------------------------------------------- chat

    @r.post("/api/chat")
    async def chat(request: Request):
        body = await request.json()
        model = body.get("model", "")
        messages = body.get("messages")
        if not isinstance(messages, list) or not messages:
            raise HTTPException(status_code=400,
                                detail="messages must be a non-empty array")
        total = 0
        for m in messages:             # Joi parity (ollama.ts:60-117): a
            if not isinstance(m, dict) or \
                    not isinstance(m.get("role", ""), str) or \
                    not isinstance(m.get("content", ""), (str, list)):
                raise HTTPException(   # junk message must 400 here, not
                    status_code=400,   # crash inside the worker
                    detail="each message needs a string role and "
                           "string/array content")
            c = m.get("content", "")
            total += len(c) if isinstance(c, str) else \
                sum(len(str(p)) for p in c)
        if total > 100_000:            # same cap the generate path has
            raise HTTPException(status_code=400,
                                detail="messages too long (max 100kB)")
        _validate_model(registry, model)
        stream = bool(body.get("stream", True))
        opts = _clean_options(body.get("options"))
        fmt = body.get("format")
        if fmt == "json" or isinstance(fmt, dict):
            opts["format"] = fmt       # dict = JSON schema (enforced)
        # tools: render into a system instruction; parse the JSON reply
        # back into message.tool_calls (arguments as OBJECT per Ollama)
        fns = tools_mod.extract_functions(body.get("tools"))
        if fns:
            messages = [tools_mod.system_message(fns)] + list(messages)
        req = InferenceRequest(
            model=model, messages=messages, request_type="chat",
            stream=stream, timeout_ms=300_000, options=opts,
            metadata={"think": True} if body.get("think") else None)

        def chunk_json(c):
            msg = {"role": "assistant", "content": c.get("response", "")}
            if c.get("thinking"):
                msg["thinking"] = c["thinking"]
            return {"model": model, "created_at": _now_iso(),
                    "message": msg,
                    "done": False}

        def final_json(res):
            msg = {"role": "assistant", "content": ""}
            if res.get("thinking") is not None:
                msg["thinking"] = ""
            if fns:
                # Ollama emits tool_calls on the final chunk of a stream
                calls = tools_mod.parse_calls(res.get("response", "") or "")
                if calls:
                    msg["tool_calls"] = tools_mod.to_ollama(calls)
            return {"model": model, "created_at": _now_iso(),
                    "message": msg,
                    "done": True,
                    "done_reason": res.get("done_reason", "stop"),
                    **_timings(res)}

        if stream:
            return _stream_ndjson(scheduler, req, chunk_json, final_json)
        res = await scheduler.submit_and_wait(req)
        if res.get("error"):
            raise HTTPException(status_code=500, detail=res["error"])
        msg = res.get("message") or \
            {"role": "assistant", "content": res.get("response", "")}
        if fns and msg.get("content"):
            calls = tools_mod.parse_calls(msg["content"])
            if calls:
                msg = {"role": "assistant", "content": "",
                       "tool_calls": tools_mod.to_ollama(calls)}
        return JSONResponse({
            "model": model, "created_at": _now_iso(),
            "message": msg,
            "done": True, "done_reason": res.get("done_reason", "stop"),
            **_timings(res)})

    # ----------------------------------------------------------------- tags

    @r.get("/api/tags")
    async def tags():
        models = registry.get_all_models()
        out = []
        for name, n_workers in sorted(models.items()):
            out.append({
                "name": name, "model": name,
                "modified_at": _now_iso(),
                "size": 0, "digest": "",
                "details": {"parent_model": "", "format": "safetensors",
                            "family": "llama", "families": ["llama"],
                            "parameter_size": "", "quantization_level": "BF16"},
                "gridllm_metadata": {"num_workers_with_model": n_workers},
            })
        return JSONResponse({"models": out})

    # ------------------------------------------------- version / show / ps
    # (beyond reference parity: the reference proxied only
    # generate/chat/tags/embed — SURVEY.md §2.2-S5 — but Ollama SDKs probe
    # these on connect, so a drop-in gateway should answer them)

    @r.get("/api/version")
    async def version():
        from gridllm_amd import __version__
        return JSONResponse({"version": __version__})

    @r.post("/api/show")
    async def show(request: Request):
        body = await request.json()
        model = body.get("model") or body.get("name") or ""
        _validate_model(registry, model)
        try:
            from gridllm_amd.models.registry import get_spec
            spec = get_spec(model)
            info = {
                "general.architecture": "llama",
                "general.parameter_count": spec.params,
                "llama.block_count": spec.layers,
                "llama.embedding_length": spec.hidden,
                "llama.attention.head_count": spec.heads,
                "llama.attention.head_count_kv": spec.kv_heads,
                "llama.feed_forward_length": spec.intermediate,
                "llama.context_length": spec.max_seq_len,
                "llama.vocab_size": spec.vocab,
                "llama.rope.freq_base": spec.rope_theta,
            }
        except KeyError:
            info = {}
        return JSONResponse({
            "modelfile": "", "parameters": "", "template": "",
            "details": {"parent_model": "", "format": "safetensors",
                        "family": "llama", "families": ["llama"],
                        "parameter_size": "", "quantization_level": "BF16"},
            "model_info": info,
            "capabilities": ["completion", "chat", "embedding"],
        })

    @r.get("/api/ps")
    async def ps():
        out = []
        for w in registry.all_workers():
            for name in w.models:
                out.append({
                    "name": name, "model": name,
                    "size": 0, "digest": "",
                    "details": {"family": "llama",
                                "quantization_level": "BF16"},
                    "expires_at": _now_iso(),
                    "size_vram": (w.gpu or {}).get("hbm_used_mb", 0)
                    * 1024 * 1024,
                    "gridllm_metadata": {"worker_id": w.id,
                                         "status": w.status},
                })
        return JSONResponse({"models": out})

    # ---------------------------------------------------------------- embed

    @r.post("/api/embed")
    async def embed(request: Request):
        body = await request.json()
        model = body.get("model", "")
        inp = body.get("input")
        if inp is None:
            raise HTTPException(status_code=400, detail="input is required")
        if not isinstance(inp, str) and not (
                isinstance(inp, list)
                and all(isinstance(x, str) for x in inp)):
            raise HTTPException(
                status_code=400,
                detail="input must be a string or array of strings")
        inputs = [inp] if isinstance(inp, str) else list(inp)
        _validate_model(registry, model)
        req = InferenceRequest(model=model, request_type="embedding",
                               input=inputs, timeout_ms=300_000)
        res = await scheduler.submit_and_wait(req)
        if res.get("error"):
            raise HTTPException(status_code=500, detail=res["error"])
        t = _timings(res)
        return JSONResponse({
            "model": model, "embeddings": res.get("embeddings", []),
            "total_duration": t["total_duration"],
            "load_duration": t["load_duration"],
            "prompt_eval_count": t["prompt_eval_count"]})

    @r.post("/api/embeddings")
    async def embeddings_legacy(request: Request):
        body = await request.json()
        model = body.get("model", "")
        prompt = body.get("prompt")
        if prompt is None:
            raise HTTPException(status_code=400, detail="prompt is required")
        if not isinstance(prompt, str):
            raise HTTPException(status_code=400,
                                detail="prompt must be a string")
        _validate_model(registry, model)
        req = InferenceRequest(model=model, request_type="embedding",
                               input=[prompt], timeout_ms=300_000)
        res = await scheduler.submit_and_wait(req)
        if res.get("error"):
            raise HTTPException(status_code=500, detail=res["error"])
        embs = res.get("embeddings") or [[]]
        return JSONResponse({"embedding": embs[0]})

    return r
