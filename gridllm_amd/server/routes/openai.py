"""OpenAI-compatible API (reference: server/src/routes/openai.ts).

/v1/completions, /v1/chat/completions (SSE streaming with `data: ...` frames
and a terminating `data: [DONE]`, :446-549/:695-789), /v1/models (:822-874).
Option mapping mirrors :387-411 (max_tokens -> num_predict etc.); chat
requests run the real chat path (:651-654). response_format json/json_schema
accepted and forwarded as metadata.
"""

from __future__ import annotations

import asyncio
import json
import time
import uuid

from fastapi import APIRouter, HTTPException, Request
from fastapi.responses import JSONResponse, StreamingResponse

from gridllm_amd.server import tools as tools_mod
from gridllm_amd.server.registry import WorkerRegistry
from gridllm_amd.server.scheduler import JobScheduler
from gridllm_amd.types import InferenceRequest


def _map_options(body: dict) -> dict:
    """OpenAI params -> engine sampling options (reference openai.ts:387-411)."""
    out: dict = {}
    if body.get("temperature") is not None:
        out["temperature"] = body["temperature"]
    if body.get("top_p") is not None:
        out["top_p"] = body["top_p"]
    if body.get("max_tokens") is not None:
        out["num_predict"] = body["max_tokens"]
    if body.get("max_completion_tokens") is not None:
        out["num_predict"] = body["max_completion_tokens"]
    if body.get("seed") is not None:
        out["seed"] = body["seed"]
    if body.get("stop") is not None:
        stop = body["stop"]
        if isinstance(stop, str):
            out["stop"] = [stop]
        elif isinstance(stop, list):
            out["stop"] = list(stop)
        else:
            from fastapi import HTTPException
            raise HTTPException(
                status_code=400,
                detail="stop must be a string or array of strings")
    if body.get("presence_penalty") is not None:
        out["presence_penalty"] = body["presence_penalty"]
    if body.get("frequency_penalty") is not None:
        out["frequency_penalty"] = body["frequency_penalty"]
    from gridllm_amd.types import SamplingOptions
    err = SamplingOptions.validate(out)
    if err:                        # OpenAI 400s invalid param types
        from fastapi import HTTPException
        raise HTTPException(status_code=400, detail=err)
    return out


def _usage(res: dict) -> dict:
    t = res.get("timings") or {}
    p = int(t.get("prompt_eval_count", 0))
    c = int(t.get("eval_count", 0))
    return {"prompt_tokens": p, "completion_tokens": c, "total_tokens": p + c}


def _usage_n(results: list) -> dict:
    """Usage across n samples of ONE prompt: the shared prompt counts once;
    completion tokens sum (OpenAI semantics — round-1 advisor finding)."""
    p = int((results[0].get("timings") or {}).get("prompt_eval_count", 0))
    c = sum(int((r.get("timings") or {}).get("eval_count", 0))
            for r in results)
    return {"prompt_tokens": p, "completion_tokens": c,
            "total_tokens": p + c}


def _finish_reason(res: dict) -> str:
    return {"length": "length", "cancelled": "stop"}.get(
        res.get("done_reason", "stop"), "stop")


def _convert_messages(messages: list) -> list:
    """OpenAI messages -> engine messages, incl. multimodal text/image split
    (reference openai.ts:204-243)."""
    out = []
    for m in messages:
        role = m.get("role", "user")
        content = m.get("content", "")
        if content is None:
            content = ""
        # assistant tool-call history: re-serialize the calls so the
        # conversation context survives the round trip
        if role == "assistant" and m.get("tool_calls"):
            try:
                calls = [{"name": (c.get("function") or {}).get("name"),
                          "arguments": (c.get("function") or {})
                          .get("arguments")} for c in m["tool_calls"]]
                content = (content + "\n" if content else "") + \
                    json.dumps(calls if len(calls) > 1 else calls[0])
            except Exception:  # noqa: BLE001
                pass
        images = []
        if isinstance(content, list):
            texts = []
            for part in content:
                if not isinstance(part, dict):
                    continue
                if part.get("type") == "text":
                    texts.append(part.get("text", ""))
                elif part.get("type") == "image_url":
                    url = (part.get("image_url") or {}).get("url", "")
                    if url.startswith("data:") and "," in url:
                        images.append(url.split(",", 1)[1])
            content = "\n".join(texts)
        msg = {"role": role, "content": content}
        if images:
            msg["images"] = images
        out.append(msg)
    return out


def build_router(scheduler: JobScheduler, registry: WorkerRegistry) -> APIRouter:
    r = APIRouter()

    def _validate(model: str) -> None:
        if not model:
            raise HTTPException(status_code=400, detail="model is required")
        if not registry.get_workers_by_model(model):
            raise HTTPException(status_code=404,
                                detail=f"model '{model}' not found")

    def _sse(scheduler: JobScheduler, req: InferenceRequest, make_chunk,
             make_final, buffered_finals=None):
        """buffered_finals: when set (tool-call streams), token deltas are
        withheld and the callback maps the FINAL result to the list of SSE
        payloads — a tool call can only be recognized on the complete
        text, matching OpenAI's tool_call-delta framing."""
        async def gen():
            q: asyncio.Queue = asyncio.Queue()

            def on_chunk(c: dict) -> None:
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
                        if item.get("response") and buffered_finals is None:
                            yield f"data: {json.dumps(make_chunk(item))}\n\n"
                    elif kind == "final":
                        if not item.get("error"):
                            if buffered_finals is not None:
                                for payload in buffered_finals(item):
                                    yield f"data: {json.dumps(payload)}\n\n"
                            else:
                                yield \
                                    f"data: {json.dumps(make_final(item))}\n\n"
                        yield "data: [DONE]\n\n"
                        break
                    else:
                        yield f"data: {json.dumps({'error': item})}\n\n"
                        yield "data: [DONE]\n\n"
                        break
            finally:
                task.cancel()
                if not task.done():
                    # client disconnected mid-stream: abort the job on the
                    # worker too, or it keeps generating to completion
                    # (round-1 advisor finding)
                    try:
                        await scheduler.cancel_job(req.id)
                    except Exception:  # noqa: BLE001
                        pass

        return StreamingResponse(gen(), media_type="text/event-stream")

    # ---------------------------------------------------------- completions

    @r.post("/v1/completions")
    async def completions(request: Request):
        body = await request.json()
        model = body.get("model", "")
        _validate(model)
        prompt = body.get("prompt", "")
        if isinstance(prompt, list):
            prompt = prompt[0] if prompt else ""
        if not isinstance(prompt, str):
            raise HTTPException(status_code=400,
                                detail="prompt must be a string or "
                                       "array of strings")
        stream = bool(body.get("stream", False))
        echo = bool(body.get("echo", False))
        cid = f"cmpl-{uuid.uuid4().hex[:24]}"
        created = int(time.time())
        req = InferenceRequest(model=model, prompt=prompt,
                               request_type="generate", stream=stream,
                               timeout_ms=300_000, options=_map_options(body))

        def make_chunk(c):
            return {"id": cid, "object": "text_completion", "created": created,
                    "model": model,
                    "choices": [{"text": c.get("response", ""), "index": 0,
                                 "logprobs": None, "finish_reason": None}]}

        def make_final(res):
            return {"id": cid, "object": "text_completion", "created": created,
                    "model": model,
                    "choices": [{"text": "", "index": 0, "logprobs": None,
                                 "finish_reason": _finish_reason(res)}],
                    "usage": _usage(res)}

        if stream:
            return _sse(scheduler, req, make_chunk, make_final)
        n = max(1, min(int(body.get("n") or 1), 16))
        reqs = [req]
        for i in range(1, n):       # n samples: distinct seeded streams
            o = dict(req.options)
            if o.get("seed") is not None:
                o["seed"] = int(o["seed"]) + i
            reqs.append(InferenceRequest(
                model=model, prompt=prompt, request_type="generate",
                stream=False, timeout_ms=300_000, options=o))
        results = await asyncio.gather(
            *(scheduler.submit_and_wait(rq) for rq in reqs))
        for res in results:
            if res.get("error"):
                raise HTTPException(status_code=500, detail=res["error"])
        choices = []
        for i, res in enumerate(results):
            text = res.get("response", "")
            if echo:
                text = prompt + text
            choices.append({"text": text, "index": i, "logprobs": None,
                            "finish_reason": _finish_reason(res)})
        return JSONResponse({
            "id": cid, "object": "text_completion", "created": created,
            "model": model, "choices": choices,
            "usage": _usage_n(results)})

    # ----------------------------------------------------- chat completions

    @r.post("/v1/chat/completions")
    async def chat_completions(request: Request):
        body = await request.json()
        model = body.get("model", "")
        _validate(model)
        messages = body.get("messages")
        if not isinstance(messages, list) or not messages:
            raise HTTPException(status_code=400,
                                detail="messages must be a non-empty array")
        for m in messages:
            if not isinstance(m, dict) or \
                    not isinstance(m.get("role", ""), str) or \
                    not isinstance(m.get("content", ""),
                                   (str, list, type(None))):
                raise HTTPException(
                    status_code=400,
                    detail="each message needs a string role and "
                           "string/array content")
        stream = bool(body.get("stream", False))
        cid = f"chatcmpl-{uuid.uuid4().hex[:24]}"
        created = int(time.time())
        options = _map_options(body)
        rf = body.get("response_format")
        if isinstance(rf, dict) and rf.get("type") == "json_object":
            options["format"] = "json"
        elif isinstance(rf, dict) and rf.get("type") == "json_schema":
            # the engine ENFORCES the schema (schema-directed byte
            # automaton, engine/schema_constraints.py)
            schema = (rf.get("json_schema") or {}).get("schema")
            options["format"] = schema if isinstance(schema, dict) \
                else "json"
        conv = _convert_messages(messages)
        # tools (reference forwarded them to Ollama's template and
        # converted returned tool_calls, openai.ts:246-304,651-654):
        # render specs into a system instruction; force format=json
        # (byte-level automaton) when the call is required
        fns = tools_mod.extract_functions(body.get("tools"))
        tool_choice = body.get("tool_choice")
        if fns and tool_choice != "none":
            conv = [tools_mod.system_message(fns, tool_choice)] + conv
            if tools_mod.forced_call(tool_choice):
                # schema-enforced call: name from the allowed set and
                # arguments valid against the function's parameters
                options["format"] = tools_mod.call_schema(fns, tool_choice)
        req = InferenceRequest(model=model,
                               messages=conv,
                               request_type="chat", stream=stream,
                               timeout_ms=300_000, options=options)

        def make_chunk(c):
            return {"id": cid, "object": "chat.completion.chunk",
                    "created": created, "model": model,
                    "choices": [{"index": 0,
                                 "delta": {"content": c.get("response", "")},
                                 "finish_reason": None}]}

        def make_final(res):
            return {"id": cid, "object": "chat.completion.chunk",
                    "created": created, "model": model,
                    "choices": [{"index": 0, "delta": {},
                                 "finish_reason": _finish_reason(res)}],
                    "usage": _usage(res)}

        if stream:
            finals = None
            if fns and tool_choice != "none":
                def finals(res):           # noqa: F811 — buffered stream
                    text = res.get("response", "") or \
                        (res.get("message") or {}).get("content", "")
                    calls = tools_mod.parse_calls(text) if text else None
                    base = {"id": cid, "object": "chat.completion.chunk",
                            "created": created, "model": model}
                    if calls:
                        deltas = [{**base, "choices": [{
                            "index": 0, "finish_reason": None,
                            "delta": {"role": "assistant",
                                      "tool_calls": [{"index": i, **c}]}}]}
                            for i, c in
                            enumerate(tools_mod.to_openai(calls))]
                        return deltas + [{**base, "choices": [{
                            "index": 0, "delta": {},
                            "finish_reason": "tool_calls"}],
                            "usage": _usage(res)}]
                    out = []
                    if text:
                        out.append({**base, "choices": [{
                            "index": 0, "finish_reason": None,
                            "delta": {"content": text}}]})
                    return out + [make_final(res)]
            return _sse(scheduler, req, make_chunk, make_final,
                        buffered_finals=finals)
        n = max(1, min(int(body.get("n") or 1), 16))
        reqs = [req]
        for i in range(1, n):       # n samples: distinct seeded streams
            o = dict(req.options)
            if o.get("seed") is not None:
                o["seed"] = int(o["seed"]) + i
            reqs.append(InferenceRequest(
                model=model, messages=conv, request_type="chat",
                stream=False, timeout_ms=300_000, options=o))
        results = await asyncio.gather(
            *(scheduler.submit_and_wait(rq) for rq in reqs))
        for res in results:
            if res.get("error"):
                raise HTTPException(status_code=500, detail=res["error"])
        choices = []
        for i, res in enumerate(results):
            msg = res.get("message") or {"role": "assistant",
                                         "content": res.get("response", "")}
            finish = _finish_reason(res)
            if fns and tool_choice != "none" and msg.get("content"):
                calls = tools_mod.parse_calls(msg["content"])
                if calls:
                    msg = {"role": "assistant", "content": None,
                           "tool_calls": tools_mod.to_openai(calls)}
                    finish = "tool_calls"
            choices.append({"index": i, "message": msg, "logprobs": None,
                            "finish_reason": finish})
        return JSONResponse({
            "id": cid, "object": "chat.completion", "created": created,
            "model": model, "choices": choices,
            "usage": _usage_n(results)})

    # --------------------------------------------------------------- models

    @r.post("/v1/embeddings")
    async def embeddings(request: Request):
        # Ollama's OpenAI compat exposes this; the reference proxied only
        # completions/chat/models — drop-in value-add
        body = await request.json()
        model = body.get("model", "")
        _validate(model)
        raw = body.get("input", "")
        inputs = raw if isinstance(raw, list) else [raw]
        inputs = [str(x) for x in inputs]
        req = InferenceRequest(model=model, request_type="embedding",
                               input=inputs, timeout_ms=300_000)
        res = await scheduler.submit_and_wait(req)
        if res.get("error"):
            raise HTTPException(status_code=500, detail=res["error"])
        embs = res.get("embeddings", [])
        return JSONResponse({
            "object": "list",
            "data": [{"object": "embedding", "index": i, "embedding": e}
                     for i, e in enumerate(embs)],
            "model": model,
            "usage": {"prompt_tokens": 0, "total_tokens": 0},
        })

    @r.get("/v1/models")
    async def models():
        out = [{"id": name, "object": "model", "created": int(time.time()),
                "owned_by": "gridllm"}
               for name in sorted(registry.get_all_models())]
        return JSONResponse({"object": "list", "data": out})

    return r
