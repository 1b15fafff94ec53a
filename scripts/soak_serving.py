"""Serving soak: many concurrent requests through the full mesh with the
continuous-batching engine (stability under load)."""
import asyncio
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from crowdllama_amd.config import Config
from crowdllama_amd.engine.batching import BatchingHipEngine
from crowdllama_amd.mesh.dhtnode import DHTServer
from crowdllama_amd.mesh.gateway import Gateway
from crowdllama_amd.mesh.peer import Peer
from crowdllama_amd.models import synth_path


async def main():
    import aiohttp
    path = synth_path("testllama", scheme="q4_k_m", mode="exact", seed=7)
    eng = BatchingHipEngine("testllama", path, batch=8, max_seq=256)
    def mk(c):
        return Config(test_mode=True, listen_host="127.0.0.1",
                      key_path=f"/tmp/soak-{c}.key")
    dht = DHTServer(mk("dht"), "CLADHT")
    port = await dht.start("127.0.0.1", 0)
    wcfg = mk("worker"); wcfg.bootstrap_peers = [f"127.0.0.1:{port}"]
    worker = Peer(wcfg, worker_mode=True, engines={"testllama": eng})
    await worker.start()
    ccfg = mk("consumer"); ccfg.bootstrap_peers = [f"127.0.0.1:{port}"]
    consumer = Peer(ccfg, worker_mode=False)
    await consumer.start()
    gw = Gateway(consumer, ccfg)
    gport = await gw.start(port=0)
    while gw.find_best_worker("testllama") is None:
        await asyncio.sleep(0.1)
    N = 120
    t0 = time.time()
    ok = 0
    async with aiohttp.ClientSession() as s:
        async def one(i):
            nonlocal ok
            async with s.post(f"http://127.0.0.1:{gport}/api/chat",
                              json={"model": "testllama",
                                    "messages": [{"role": "user",
                                                  "content": f"req {i} " * (1 + i % 5)}]},
                              timeout=aiohttp.ClientTimeout(total=300)) as r:
                body = await r.json()
                assert r.status == 200, body
                assert body["done"] is True
                ok += 1
        await asyncio.gather(*[one(i) for i in range(N)])
    dt = time.time() - t0
    print(f"soak: {ok}/{N} requests ok in {dt:.1f}s "
          f"({ok/dt:.1f} req/s through the mesh, batch=8)")
    st = worker.peer_manager.get_peer_statistics()
    print(f"served={worker.requests_served} peers={st}")
    await gw.stop(); await consumer.stop(); await worker.stop(); await dht.stop()

asyncio.run(main())
