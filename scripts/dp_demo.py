"""DP load-balancing demo on one box: DHT + TWO HIP workers (one GPU,
separate identities) + gateway; fires concurrent chats and reports how the
scheduler spread them (reference behavior: score = throughput/(1+load),
manager.go:338-387 — here with measured throughput and live load)."""
import asyncio
import collections
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


async def main():
    import aiohttp
    from crowdllama_amd.config import Config
    from crowdllama_amd.engine.hip_engine import HipEngine
    from crowdllama_amd.mesh.dhtnode import DHTServer
    from crowdllama_amd.mesh.gateway import Gateway
    from crowdllama_amd.mesh.peer import Peer
    from crowdllama_amd.models import synth_path

    path = synth_path("tinyllama", scheme="q8_0", mode="fast")

    def mk(c):
        return Config(test_mode=True, listen_host="127.0.0.1",
                      key_path=f"/tmp/cla-dp-{c}.key")

    dht = DHTServer(mk("dht"), "CLADHT")
    port = await dht.start("127.0.0.1", 0)
    boot = [f"127.0.0.1:{port}"]
    workers = []
    for i in range(2):
        wcfg = mk(f"worker{i}")
        wcfg.bootstrap_peers = boot
        w = Peer(wcfg, worker_mode=True,
                 engines={"tinyllama": HipEngine("tinyllama", path,
                                                 max_seq=512)})
        await w.start()
        workers.append(w)
    ccfg = mk("consumer")
    ccfg.bootstrap_peers = boot
    consumer = Peer(ccfg, worker_mode=False)
    await consumer.start()
    gw = Gateway(consumer, ccfg)
    gport = await gw.start(port=0)

    deadline = time.time() + 30
    while len({r.peer_id for r in consumer.peer_manager.get_healthy_peers()
               if r.worker_mode}) < 2:
        assert time.time() < deadline, "workers not discovered"
        await asyncio.sleep(0.2)

    served = collections.Counter()
    t0 = time.time()

    async def one(sess, i):
        async with sess.post(f"http://127.0.0.1:{gport}/api/chat",
                             json={"model": "tinyllama",
                                   "messages": [{"role": "user",
                                                 "content": f"req {i}"}]}) as r:
            body = await r.json()
            assert r.status == 200, body
            served[body["worker_id"]] += 1

    async with aiohttp.ClientSession() as sess:
        for wave in range(10):
            await asyncio.gather(*[one(sess, wave * 8 + i) for i in range(8)])
    dt = time.time() - t0
    ids = {w.peer_id: f"worker{i}" for i, w in enumerate(workers)}
    print(f"80 requests in {dt:.1f}s ({80 / dt:.1f} req/s) across "
          f"{len(served)} workers:")
    for pid, n in served.items():
        print(f"  {ids.get(pid, pid)}: {n} requests "
              f"(served={next(w.requests_served for w in workers if w.peer_id == pid)})")
    assert len(served) == 2, "expected both workers to take traffic"

    await gw.stop()
    await consumer.stop()
    for w in workers:
        await w.stop()
    await dht.stop()
    print("dp demo ok")


if __name__ == "__main__":
    asyncio.run(main())
