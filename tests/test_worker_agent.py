"""WorkerAgent end-to-end against the real (tiny, CPU) engine via the
scheduler — registration, generate, chat, embedding, streaming, cancellation
(the reference never honored cancellation: WorkerClientService.ts:483-491)."""

import asyncio

import pytest

from gridllm_amd.bus.memory import MemoryBus, MemoryBusCore
from gridllm_amd.config import EngineConfig, ServerConfig, WorkerConfig
from gridllm_amd.engine import InferenceEngine
from gridllm_amd.server.registry import WorkerRegistry
from gridllm_amd.server.scheduler import JobScheduler
from gridllm_amd.types import InferenceRequest
from gridllm_amd.worker.agent import WorkerAgent

FAST = ServerConfig(worker_heartbeat_timeout_ms=500,
                    worker_cleanup_interval_ms=100,
                    job_timeout_ms=10000, job_retry_attempts=1,
                    job_retry_delay_ms=50)

_ENGINE = None


def get_engine():
    global _ENGINE
    if _ENGINE is None:
        _ENGINE = InferenceEngine(EngineConfig(
            model="tiny", device="cpu", dtype="fp32", max_seq_len=128,
            max_batch_size=4, use_hipgraph=False))
    return _ENGINE


async def make_world(core):
    bus_s = MemoryBus(core)
    await bus_s.connect()
    reg = WorkerRegistry(bus_s, FAST)
    sched = JobScheduler(bus_s, reg, FAST)
    await reg.initialize()
    await sched.initialize()
    bus_w = MemoryBus(core)
    await bus_w.connect()
    agent = WorkerAgent(bus_w, get_engine(),
                        WorkerConfig(worker_id="w-test", max_concurrent_tasks=2,
                                     heartbeat_interval_ms=200))
    await agent.start()
    await asyncio.sleep(0.05)
    return reg, sched, agent


async def teardown(reg, sched, agent):
    await agent.stop()
    await sched.shutdown()
    await reg.shutdown()


@pytest.mark.asyncio
async def test_agent_registers_with_real_capabilities():
    core = MemoryBusCore()
    reg, sched, agent = await make_world(core)
    info = reg.get("w-test")
    assert info is not None
    assert "tiny" in info.models
    assert info.max_concurrent_jobs == 2
    await teardown(reg, sched, agent)


@pytest.mark.asyncio
async def test_generate_roundtrip():
    core = MemoryBusCore()
    reg, sched, agent = await make_world(core)
    req = InferenceRequest(model="tiny", prompt="hello",
                           options={"temperature": 0.0, "num_predict": 5},
                           timeout_ms=30000)
    res = await asyncio.wait_for(sched.submit_and_wait(req), 30)
    assert res.get("error") is None
    assert res["done_reason"] in ("stop", "length")
    assert res["timings"]["eval_count"] >= 1
    assert res["timings"]["prompt_eval_count"] >= 1
    await teardown(reg, sched, agent)


@pytest.mark.asyncio
async def test_streaming_roundtrip():
    core = MemoryBusCore()
    reg, sched, agent = await make_world(core)
    chunks = []
    req = InferenceRequest(model="tiny", prompt="stream",
                           options={"temperature": 0.0, "num_predict": 6},
                           stream=True, timeout_ms=30000)
    res = await asyncio.wait_for(
        sched.submit_streaming(req, lambda c: chunks.append(c)), 30)
    await asyncio.sleep(0.1)
    assert res.get("error") is None
    streamed = "".join(c["response"] for c in chunks)
    assert streamed == res["response"]
    assert any(c["done"] for c in chunks)
    await teardown(reg, sched, agent)


@pytest.mark.asyncio
async def test_chat_roundtrip():
    core = MemoryBusCore()
    reg, sched, agent = await make_world(core)
    req = InferenceRequest(model="tiny", request_type="chat",
                           messages=[{"role": "user", "content": "hi"}],
                           options={"temperature": 0.0, "num_predict": 4},
                           timeout_ms=30000)
    res = await asyncio.wait_for(sched.submit_and_wait(req), 30)
    assert res.get("error") is None
    assert res["message"]["role"] == "assistant"
    await teardown(reg, sched, agent)


@pytest.mark.asyncio
async def test_embedding_roundtrip():
    core = MemoryBusCore()
    reg, sched, agent = await make_world(core)
    req = InferenceRequest(model="tiny", request_type="embedding",
                           input=["alpha", "beta"], timeout_ms=30000)
    res = await asyncio.wait_for(sched.submit_and_wait(req), 30)
    assert res.get("error") is None
    assert len(res["embeddings"]) == 2
    await teardown(reg, sched, agent)


@pytest.mark.asyncio
async def test_cancellation_honored():
    """cancel_job must abort in-flight generation on the worker."""
    core = MemoryBusCore()
    reg, sched, agent = await make_world(core)
    req = InferenceRequest(model="tiny", prompt="long",
                           options={"temperature": 0.0, "num_predict": 100000},
                           timeout_ms=60000)
    task = asyncio.create_task(sched.submit_and_wait(req))
    # wait until the job is actually running on the worker
    for _ in range(100):
        await asyncio.sleep(0.02)
        if req.id in agent._cancel_events:
            break
    assert req.id in agent._cancel_events
    ok = await sched.cancel_job(req.id)
    assert ok
    res = await asyncio.wait_for(task, 30)
    assert res.get("error") == "cancelled"
    # worker must come back idle (generation aborted, slot freed)
    for _ in range(200):
        await asyncio.sleep(0.02)
        if agent.current_jobs == 0:
            break
    assert agent.current_jobs == 0
    assert get_engine().model.cache.n_free == \
        get_engine().cfg.max_batch_size
    await teardown(reg, sched, agent)


@pytest.mark.asyncio
async def test_concurrent_jobs_batched():
    """Two concurrent jobs coalesce into one engine wave and both finish."""
    core = MemoryBusCore()
    reg, sched, agent = await make_world(core)
    reqs = [InferenceRequest(model="tiny", prompt=p,
                             options={"temperature": 0.0, "num_predict": 5},
                             timeout_ms=30000) for p in ("a", "b")]
    results = await asyncio.wait_for(
        asyncio.gather(*(sched.submit_and_wait(r) for r in reqs)), 30)
    assert all(r.get("error") is None for r in results)
    await teardown(reg, sched, agent)


@pytest.mark.asyncio
async def test_pause_resume_drain_verbs():
    """Maintenance verbs adopted from the reference's dead broker path:
    paused workers reject assignments and leave the schedulable pool;
    drain finishes in-flight work then unregisters."""
    from gridllm_amd.types import Channels
    core = MemoryBusCore()
    reg, sched, agent = await make_world(core)
    bus = MemoryBus(core)
    await bus.connect()
    wid = agent.worker_id
    try:
        await bus.publish(Channels.worker_control(wid), {"type": "pause"})
        await asyncio.sleep(0.1)
        assert agent.paused
        assert agent._info().status == "paused"
        assert not reg.get_available_workers_by_model("tiny")

        await bus.publish(Channels.worker_control(wid), {"type": "resume"})
        await asyncio.sleep(0.1)
        assert not agent.paused
        req = InferenceRequest(model="tiny", prompt="after resume",
                               options={"temperature": 0.0, "num_predict": 2})
        res = await asyncio.wait_for(sched.submit_and_wait(req), 30)
        assert res.get("response") is not None

        events = []

        async def on_unreg(ch, p):
            events.append(p)
        await bus.subscribe(Channels.WORKER_UNREGISTERED, on_unreg)
        await bus.publish(Channels.BROADCAST, {"type": "drain"})
        await asyncio.sleep(0.3)
        assert agent.draining
        assert any(e.get("workerId") == wid for e in events)
    finally:
        await teardown(reg, sched, agent)


@pytest.mark.asyncio
async def test_worker_health_app():
    """Worker-local health endpoints (reference client/src/routes/health.ts)
    serve status + GPU telemetry for the live agent."""
    from httpx import ASGITransport, AsyncClient
    from gridllm_amd.worker.main import build_health_app
    core = MemoryBusCore()
    reg, sched, agent = await make_world(core)
    try:
        app = build_health_app(agent)
        async with AsyncClient(transport=ASGITransport(app=app),
                               base_url="http://t") as c:
            r = await c.get("/health")
            assert r.status_code == 200 and r.json()["status"] == "healthy"
            r = await c.get("/health/ready")
            assert r.json()["status"] == "ready"
            r = await c.get("/worker/status")
            body = r.json()
            assert body["id"] == agent.worker_id
            assert "gpu" in body
            r = await c.get("/health/system")
            assert "engine" in r.json()
    finally:
        await teardown(reg, sched, agent)


@pytest.mark.asyncio
async def test_multi_model_worker():
    """ONE worker serving several models (the reference advertised Ollama's
    whole tag list): jobs dispatch to the engine matching req.model, and
    aliases are advertised so any Ollama-style spelling routes."""
    from gridllm_amd.models.registry import ModelSpec, register_spec
    register_spec(ModelSpec("tiny2", hidden=128, layers=1, heads=2,
                            kv_heads=1, head_dim=64, intermediate=256,
                            vocab=300, max_seq_len=128),
                  "tiny2:latest")
    core = MemoryBusCore()
    bus_s = MemoryBus(core)
    await bus_s.connect()
    reg = WorkerRegistry(bus_s, FAST)
    sched = JobScheduler(bus_s, reg, FAST)
    await reg.initialize()
    await sched.initialize()
    bus_w = MemoryBus(core)
    await bus_w.connect()
    e2 = InferenceEngine(EngineConfig(
        model="tiny2", device="cpu", dtype="fp32", max_seq_len=64,
        max_batch_size=2, use_hipgraph=False))
    agent = WorkerAgent(bus_w, get_engine(),
                        WorkerConfig(worker_id="w-multi",
                                     max_concurrent_tasks=4,
                                     heartbeat_interval_ms=500),
                        extra_engines=[e2])
    await agent.start()
    await asyncio.sleep(0.05)
    try:
        info = reg.get("w-multi")
        assert "tiny" in info.models and "tiny2" in info.models
        assert "tiny2:latest" in info.models        # alias advertised
        r1 = await asyncio.wait_for(sched.submit_and_wait(
            InferenceRequest(model="tiny", prompt="a",
                             options={"temperature": 0.0,
                                      "num_predict": 3})), 30)
        r2 = await asyncio.wait_for(sched.submit_and_wait(
            InferenceRequest(model="tiny2:latest", prompt="b",
                             options={"temperature": 0.0,
                                      "num_predict": 3})), 30)
        assert r1.get("response") is not None
        assert r2.get("response") is not None and not r2.get("error")
    finally:
        await teardown(reg, sched, agent)


@pytest.mark.asyncio
async def test_streaming_incremental_chunks():
    """Guards the adaptive flush cadence: a steady stream must flush
    MID-generation (several chunks), not accumulate until done — the
    full/fast-cycle gating in _flush_loop skips steady streams on fast
    cycles, and a bug there would collapse the stream to 1-2 chunks."""
    core = MemoryBusCore()
    reg, sched, agent = await make_world(core)
    chunks = []
    req = InferenceRequest(model="tiny", prompt="incremental chunks probe",
                           options={"temperature": 0.8, "seed": 7,
                                    "num_predict": 40},
                           stream=True, timeout_ms=30000)
    res = await asyncio.wait_for(
        sched.submit_streaming(req, lambda c: chunks.append(c)), 30)
    await asyncio.sleep(0.1)
    assert res.get("error") is None
    text_chunks = [c for c in chunks if c.get("response")]
    # tiny CPU engine: ~7 ms/token * 40 tokens vs 25 ms cadence -> >=3
    assert len(text_chunks) >= 3, \
        f"stream not incremental: {len(text_chunks)} text chunks"
    assert "".join(c["response"] for c in chunks) == res["response"]
    await teardown(reg, sched, agent)


@pytest.mark.asyncio
async def test_tp_wedge_watchdog_exits_group():
    """A TP batcher stuck mid-collective with work in flight must take
    the worker down (launcher then kills the group; orphan recovery
    requeues) — otherwise the asyncio loop keeps heartbeating for a
    wedged group and the scheduler routes jobs into a black hole.
    Simulated: fake-enabled TP + stale progress stamp."""
    import time as _t
    core = MemoryBusCore()
    reg, sched, agent = await make_world(core)
    engine, batcher = next(iter(agent._by_model.values()))

    class _FakeTP:
        enabled = True
    exits = []
    real_tp = engine.tp
    agent._wedge_exit = lambda code: exits.append(code)
    agent.cfg.tp_wedge_timeout_ms = 1000
    try:
        engine.tp = _FakeTP()
        # healthy: busy but fresh stamp
        batcher.busy = True
        batcher.last_progress = _t.monotonic()
        agent._check_wedge()
        assert not exits
        # idle + stale: no exit (nothing in flight, parked collectives ok)
        batcher.busy = False
        batcher.last_progress = _t.monotonic() - 999.0
        agent._check_wedge()
        assert not exits
        # busy + stale: wedge -> exit
        batcher.busy = True
        agent._check_wedge()
        assert exits == [13]
    finally:
        engine.tp = real_tp
        batcher.busy = False
    await teardown(reg, sched, agent)


@pytest.mark.asyncio
async def test_agent_subscribes_before_registering():
    """Ordering regression: the job-channel subscription must be live
    BEFORE worker:registered is announced — the reverse order lost whole
    assignment bursts to a respawned worker (scheduler assigns the
    queued backlog the moment it sees the registration; chaos repro:
    profiles/r02_logs/r02_chaos_lost_assignments.log, 16 jobs stuck
    'active' on a live idle worker)."""
    from gridllm_amd.types import Channels
    core = MemoryBusCore()
    bus = MemoryBus(core)
    await bus.connect()
    order = []
    orig_sub, orig_pub = bus.subscribe, bus.publish

    async def sub(ch, h):
        order.append(("sub", ch))
        return await orig_sub(ch, h)

    async def pub(ch, payload):
        order.append(("pub", ch))
        return await orig_pub(ch, payload)

    bus.subscribe, bus.publish = sub, pub
    agent = WorkerAgent(bus, get_engine(),
                        WorkerConfig(worker_id="w-order",
                                     max_concurrent_tasks=1,
                                     heartbeat_interval_ms=60_000))
    await agent.start()
    try:
        job_ch = Channels.worker_job("w-order")
        sub_i = order.index(("sub", job_ch))
        reg_i = order.index(("pub", Channels.WORKER_REGISTERED))
        assert sub_i < reg_i, order
    finally:
        await agent.stop()
