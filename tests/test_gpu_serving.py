"""End-to-end serving on GPU: DHT + worker with the real HIP engine +
gateway, request through /api/chat (the full reference round trip of
SURVEY.md §3.2 with first-party compute)."""

import asyncio

import pytest

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def has_gpu():
    from crowdllama_amd.ops import get_core
    c = get_core()
    if c.device_count() == 0:
        pytest.skip("no GPU")
    return True


def test_chat_through_mesh_hip_engine(has_gpu, tmp_path):
    from crowdllama_amd.config import Config
    from crowdllama_amd.engine.hip_engine import HipEngine
    from crowdllama_amd.mesh.dhtnode import DHTServer
    from crowdllama_amd.mesh.gateway import Gateway
    from crowdllama_amd.mesh.peer import Peer
    from crowdllama_amd.models import synth_path

    path = synth_path("testllama", scheme="q4_k_m", mode="exact", seed=7)
    engine = HipEngine("testllama", path, max_seq=256)

    async def go():
        import aiohttp
        import time
        def mk(c):
            return Config(test_mode=True, listen_host="127.0.0.1",
                          key_path=str(tmp_path / f"{c}.key"))
        dht = DHTServer(mk("dht"), "CLADHT")
        dht_port = await dht.start("127.0.0.1", 0)
        boot = [f"127.0.0.1:{dht_port}"]
        wcfg = mk("worker")
        wcfg.bootstrap_peers = boot
        worker = Peer(wcfg, worker_mode=True, engines={"testllama": engine})
        await worker.start()
        ccfg = mk("consumer")
        ccfg.bootstrap_peers = boot
        consumer = Peer(ccfg, worker_mode=False)
        await consumer.start()
        gw = Gateway(consumer, ccfg)
        gw_port = await gw.start(port=0)
        try:
            deadline = time.time() + 20
            while gw.find_best_worker("testllama") is None:
                assert time.time() < deadline, "worker never discovered"
                await asyncio.sleep(0.1)
            # worker metadata reflects real device props
            res = gw.find_best_worker("testllama")
            assert "gfx" in res.gpu_model or "MI" in res.gpu_model.upper(), \
                res.gpu_model
            assert res.vram_gb > 100  # MI355X: 288 GB
            async with aiohttp.ClientSession() as s:
                async with s.post(f"http://127.0.0.1:{gw_port}/api/chat",
                                  json={"model": "testllama",
                                        "messages": [{"role": "user",
                                                      "content": "abc"}]}) as r:
                    body = await r.json()
                    assert r.status == 200, body
            assert body["done"] is True
            assert body["worker_id"] == worker.peer_id
            assert isinstance(body["message"]["content"], str)
            assert body["total_duration"] > 0
        finally:
            await gw.stop()
            await consumer.stop()
            await worker.stop()
            await dht.stop()
    asyncio.run(go())


def test_hip_engine_temperature_sampling(has_gpu):
    """Host-sampling path: temperature>0 generates via logits + re-seeding
    cur_token each step."""
    import asyncio
    from crowdllama_amd.engine.hip_engine import HipEngine
    from crowdllama_amd.models import synth_path
    path = synth_path("testllama", scheme="q4_k_m", mode="exact", seed=7)
    eng = HipEngine("testllama", path, max_seq=128)
    r_greedy = asyncio.run(eng.generate("abc", max_new_tokens=8))
    r_sampled = asyncio.run(eng.generate("abc", max_new_tokens=8,
                                         temperature=1.0))
    # random-init models may emit EOS immediately; both paths must complete
    # with a valid reason and string payload
    assert r_greedy.done_reason in ("stop", "length")
    assert r_sampled.done_reason in ("stop", "length")
    assert r_greedy.tokens_generated >= 0
    assert isinstance(r_sampled.text, str)


def test_worker_serves_two_models_one_gpu(has_gpu, tmp_path):
    """Mixed-fleet building block: one worker process serving two models
    resident on one 288 GB GPU, model-aware routing picks each."""
    import asyncio
    import time
    from crowdllama_amd.config import Config
    from crowdllama_amd.engine.hip_engine import HipEngine
    from crowdllama_amd.mesh.dhtnode import DHTServer
    from crowdllama_amd.mesh.gateway import Gateway
    from crowdllama_amd.mesh.peer import Peer
    from crowdllama_amd.models import synth_path
    from crowdllama_amd.models.presets import ModelConfig
    from crowdllama_amd.models.synth import write_synthetic_gguf

    p1 = synth_path("testllama", scheme="q4_k_m", mode="exact", seed=7)
    cfg2 = ModelConfig("testllama2", vocab_size=512, hidden_size=256,
                       n_layers=2, n_heads=2, n_kv_heads=1, ffn_hidden=512,
                       rope_theta=10000.0, max_seq_len=512)
    p2 = str(tmp_path / "m2.gguf")
    write_synthetic_gguf(p2, cfg2, scheme="q8_0", mode="exact", seed=8)
    engines = {"testllama": HipEngine("testllama", p1, max_seq=128),
               "testllama2": HipEngine("testllama2", p2, max_seq=128)}

    async def go():
        import aiohttp
        def mk(c):
            return Config(test_mode=True, listen_host="127.0.0.1",
                          key_path=str(tmp_path / f"{c}.key"))
        dht = DHTServer(mk("dht"), "CLADHT")
        port = await dht.start("127.0.0.1", 0)
        wcfg = mk("worker")
        wcfg.bootstrap_peers = [f"127.0.0.1:{port}"]
        worker = Peer(wcfg, worker_mode=True, engines=engines)
        await worker.start()
        ccfg = mk("consumer")
        ccfg.bootstrap_peers = [f"127.0.0.1:{port}"]
        consumer = Peer(ccfg, worker_mode=False)
        await consumer.start()
        gw = Gateway(consumer, ccfg)
        gport = await gw.start(port=0)
        try:
            deadline = time.time() + 20
            while gw.find_best_worker("testllama2") is None:
                assert time.time() < deadline
                await asyncio.sleep(0.1)
            async with aiohttp.ClientSession() as s:
                for model in ("testllama", "testllama2"):
                    async with s.post(f"http://127.0.0.1:{gport}/api/chat",
                                      json={"model": model,
                                            "messages": [{"role": "user",
                                                          "content": "x"}]}) as r:
                        body = await r.json()
                        assert r.status == 200, body
                        assert body["model"] == model
        finally:
            await gw.stop()
            await consumer.stop()
            await worker.stop()
            await dht.stop()
    asyncio.run(go())


def test_continuous_batching_engine(has_gpu):
    """Concurrent requests share decode slots; results match the
    single-request engine (greedy, independent KV)."""
    import asyncio
    from crowdllama_amd.engine.batching import BatchingHipEngine
    from crowdllama_amd.engine.hip_engine import HipEngine
    from crowdllama_amd.models import synth_path
    path = synth_path("testllama", scheme="q4_k_m", mode="exact", seed=7)
    single = HipEngine("testllama", path, max_seq=128)
    batched = BatchingHipEngine("testllama", path, batch=2, max_seq=128)

    async def go():
        prompts = ["abc", "hello there", "xyz", "abc"]
        want = [await single.generate(p, max_new_tokens=6) for p in prompts]
        got = await asyncio.gather(
            *[batched.generate(p, max_new_tokens=6) for p in prompts])
        for w, g in zip(want, got):
            assert g.text == w.text, (g.text, w.text)
        await batched.close()
    asyncio.run(go())


def test_hip_engine_streaming_matches_generate(has_gpu):
    """generate_stream deltas concatenate to exactly the non-streamed greedy
    output (same engine, same KV discipline)."""
    import asyncio
    from crowdllama_amd.engine.hip_engine import HipEngine
    from crowdllama_amd.models import synth_path
    path = synth_path("testllama", scheme="q4_k_m", mode="exact", seed=7)
    eng = HipEngine("testllama", path, max_seq=128)

    async def go():
        want = await eng.generate("abc def", max_new_tokens=12)
        chunks = []
        async for c in eng.generate_stream("abc def", max_new_tokens=12):
            chunks.append(c)
        got = "".join(c.text for c in chunks)
        assert got == want.text, (got, want.text)
        assert chunks[-1].done_reason in ("stop", "length")
        assert all(not c.done_reason for c in chunks[:-1])
    asyncio.run(go())


def test_batching_engine_streaming(has_gpu):
    """Streaming under continuous batching: two concurrent streamed requests
    share decode strides; each stream reassembles its own text."""
    import asyncio
    from crowdllama_amd.engine.batching import BatchingHipEngine
    from crowdllama_amd.engine.hip_engine import HipEngine
    from crowdllama_amd.models import synth_path
    path = synth_path("testllama", scheme="q4_k_m", mode="exact", seed=7)
    single = HipEngine("testllama", path, max_seq=128)
    batched = BatchingHipEngine("testllama", path, batch=2, max_seq=128)

    async def collect(prompt):
        out = []
        async for c in batched.generate_stream(prompt, max_new_tokens=10):
            out.append(c.text)
        return "".join(out)

    async def go():
        w1 = await single.generate("abc", max_new_tokens=10)
        w2 = await single.generate("hello there", max_new_tokens=10)
        g1, g2 = await asyncio.gather(collect("abc"),
                                      collect("hello there"))
        assert g1 == w1.text, (g1, w1.text)
        assert g2 == w2.text, (g2, w2.text)
        await batched.close()
    asyncio.run(go())


def test_streaming_through_mesh_gpu(has_gpu, tmp_path):
    """stream=true through DHT + worker(HIP) + gateway: NDJSON chunks
    reassemble to the full response."""
    import json as _json
    import time
    from crowdllama_amd.config import Config
    from crowdllama_amd.engine.hip_engine import HipEngine
    from crowdllama_amd.mesh.dhtnode import DHTServer
    from crowdllama_amd.mesh.gateway import Gateway
    from crowdllama_amd.mesh.peer import Peer
    from crowdllama_amd.models import synth_path

    path = synth_path("testllama", scheme="q4_k_m", mode="exact", seed=7)
    engine = HipEngine("testllama", path, max_seq=128)

    async def go():
        import aiohttp
        def mk(c):
            return Config(test_mode=True, listen_host="127.0.0.1",
                          key_path=str(tmp_path / f"{c}.key"))
        dht = DHTServer(mk("dht"), "CLADHT")
        dht_port = await dht.start("127.0.0.1", 0)
        boot = [f"127.0.0.1:{dht_port}"]
        wcfg = mk("worker")
        wcfg.bootstrap_peers = boot
        worker = Peer(wcfg, worker_mode=True, engines={"testllama": engine})
        await worker.start()
        ccfg = mk("consumer")
        ccfg.bootstrap_peers = boot
        consumer = Peer(ccfg, worker_mode=False)
        await consumer.start()
        gw = Gateway(consumer, ccfg)
        gw_port = await gw.start(port=0)
        try:
            deadline = time.time() + 20
            while gw.find_best_worker("testllama") is None:
                assert time.time() < deadline
                await asyncio.sleep(0.1)
            async with aiohttp.ClientSession() as s:
                # non-streamed reference through the same mesh path (same
                # worker-side token budget)
                async with s.post(f"http://127.0.0.1:{gw_port}/api/chat",
                                  json={"model": "testllama",
                                        "messages": [{"role": "user",
                                                      "content": "abc"}]}) as r:
                    ref = await r.json()
                    assert r.status == 200, ref
                async with s.post(f"http://127.0.0.1:{gw_port}/api/chat",
                                  json={"model": "testllama", "stream": True,
                                        "messages": [{"role": "user",
                                                      "content": "abc"}]}) as r:
                    assert r.status == 200
                    lines = [_json.loads(ln) async for ln in r.content
                             if ln.strip()]
            text = "".join(ln["message"]["content"] for ln in lines)
            want = ref["message"]["content"]
            assert text == want, (text, want)
            assert lines[-1]["done"] is True
        finally:
            await gw.stop()
            await consumer.stop()
            await worker.stop()
            await dht.stop()
    asyncio.run(go())
