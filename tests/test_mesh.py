"""Mesh component tests on loopback: rendezvous server, discovery,
peer manager scheduling/health (reference parity: dht_test.go +
peermanager semantics)."""

import asyncio
import time

import pytest

from crowdllama_amd.config import Config
from crowdllama_amd.mesh.dhtnode import DHTServer
from crowdllama_amd.keys import identity_from_seed
from crowdllama_amd.mesh.discovery import Discovery, RendezvousClient

_TEST_ID = identity_from_seed(b"t" * 32)  # throwaway mesh-test identity
from crowdllama_amd.mesh.peermanager import PeerManager
from crowdllama_amd.mesh.resource import Resource


def run(coro):
    return asyncio.run(coro)


@pytest.fixture()
def cfg():
    c = Config(test_mode=True, listen_host="127.0.0.1")
    return c


def test_dht_server_start_stop(cfg):
    async def go():
        srv = DHTServer(cfg, "CLADHT")
        port = await srv.start("127.0.0.1", 0)
        assert port > 0
        cli = RendezvousClient(f"127.0.0.1:{port}", _TEST_ID)
        assert await cli.ping()
        await cli.close()
        await srv.stop()
    run(go())


def test_provide_and_find(cfg):
    async def go():
        srv = DHTServer(cfg, "CLADHT")
        port = await srv.start("127.0.0.1", 0)
        cli = RendezvousClient(f"127.0.0.1:{port}", _TEST_ID)
        me = _TEST_ID.peer_id
        # a peer may only advertise its own authenticated identity
        assert not await cli.provide("CLAFORGED", ["127.0.0.1:9"])
        ok = await cli.provide(me, ["127.0.0.1:5001"])
        assert ok
        provs = await cli.find_providers()
        assert any(p["peer_id"] == me for p in provs)
        addrs = await cli.find_peer(me)
        assert addrs == ["127.0.0.1:5001"]
        assert await cli.find_peer("CLANOBODY") is None
        # model namespace
        await cli.provide(me, ["127.0.0.1:5001"],
                          ns="crowdllama-ns/model/llama3-8b")
        provs = await cli.find_providers(ns="crowdllama-ns/model/llama3-8b")
        assert len(provs) == 1
        await cli.close()
        await srv.stop()
    run(go())


def test_provider_limit(cfg):
    async def go():
        srv = DHTServer(cfg, "CLADHT")
        port = await srv.start("127.0.0.1", 0)
        ids = [identity_from_seed(bytes([i]) * 32) for i in range(15)]
        clis = [RendezvousClient(f"127.0.0.1:{port}", ident)
                for ident in ids]
        for i, cli in enumerate(clis):
            await cli.provide(ids[i].peer_id, [f"127.0.0.1:{5000 + i}"])
        cli = clis[0]
        provs = await cli.find_providers(limit=10)
        assert len(provs) == 10  # reference: FindProvidersAsync(cid, 10)
        for c in clis:
            await c.close()
        await srv.stop()
    run(go())


def _mk_resource(pid, model="m1", thr=100.0, load=0.0, worker=True):
    r = Resource(peer_id=pid, supported_models=[model],
                 tokens_throughput=thr, load=load, worker_mode=worker,
                 addrs=["127.0.0.1:1"])
    r.touch()
    return r


def test_find_best_worker_scoring(cfg):
    """Scheduler maximizes throughput/(1+load) (manager.go:338-387)."""
    async def go():
        disco = Discovery([], _TEST_ID, log=None)
        pm = PeerManager(disco, cfg.intervals)
        await pm.add_or_update_peer(_mk_resource("A", thr=100, load=0.0))
        await pm.add_or_update_peer(_mk_resource("B", thr=300, load=2.0))
        await pm.add_or_update_peer(_mk_resource("C", thr=150, load=0.2))
        # scores: A=100, B=100, C=125 -> C
        best = pm.find_best_worker("m1")
        assert best.peer_id == "C"
        # model filter
        await pm.add_or_update_peer(_mk_resource("D", model="m2", thr=999))
        assert pm.find_best_worker("m1").peer_id == "C"
        assert pm.find_best_worker("m2").peer_id == "D"
        # consumers never selected
        await pm.add_or_update_peer(_mk_resource("E", thr=9999, worker=False))
        assert pm.find_best_worker("m1").peer_id == "C"
        # no worker for unknown model
        assert pm.find_best_worker("nope") is None
    run(go())


def test_tombstones_prevent_readd(cfg):
    async def go():
        disco = Discovery([], _TEST_ID, log=None)
        pm = PeerManager(disco, cfg.intervals)
        await pm.add_or_update_peer(_mk_resource("A"))
        await pm.remove_peer("A")
        assert "A" not in pm.peers
        await pm.add_or_update_peer(_mk_resource("A"))
        assert "A" not in pm.peers  # tombstoned
        # expire the tombstone manually and re-add
        pm.recently_removed["A"] = time.time() - cfg.intervals.tombstone - 1
        await pm.add_or_update_peer(_mk_resource("A"))
        assert "A" in pm.peers
    run(go())


def test_stale_cleanup(cfg):
    async def go():
        disco = Discovery([], _TEST_ID, log=None)
        pm = PeerManager(disco, cfg.intervals)
        await pm.add_or_update_peer(_mk_resource("A"))
        pm.peers["A"].last_seen = time.time() - cfg.intervals.stale_timeout - 1
        await pm.start()
        try:
            deadline = time.time() + 10
            while "A" in pm.peers and time.time() < deadline:
                await asyncio.sleep(0.1)
            assert "A" not in pm.peers
        finally:
            await pm.stop()
    run(go())


def test_self_never_added(cfg):
    async def go():
        disco = Discovery([], _TEST_ID, log=None)
        pm = PeerManager(disco, cfg.intervals, self_id="ME")
        await pm.add_or_update_peer(_mk_resource("ME"))
        assert "ME" not in pm.peers
    run(go())


def test_provider_ttl_expiry(cfg, monkeypatch):
    """Provider records expire without re-provide (reference: the 1 s
    advertise loop keeps records alive; dead peers age out)."""
    async def go():
        monkeypatch.setattr(DHTServer, "PROVIDER_TTL", 0.6)
        srv = DHTServer(cfg, "CLADHT")
        port = await srv.start("127.0.0.1", 0)
        cl = RendezvousClient(f"127.0.0.1:{port}", _TEST_ID)
        try:
            await cl.provide(_TEST_ID.peer_id, ["127.0.0.1:1111"], "ns")
            assert len(await cl.find_providers("ns")) == 1
            await asyncio.sleep(1.3)
            assert await cl.find_providers("ns") == []
        finally:
            await cl.close()
            await srv.stop()
    run(go())


def test_request_failure_marks_unhealthy(cfg):
    """Gateway-reported request failures accumulate into the same counter
    the health checker uses; max_failed_attempts flips is_healthy."""
    async def go():
        disc = Discovery([], _TEST_ID)
        pm = PeerManager(disc, cfg.intervals, self_id="CLAME")
        r = Resource(peer_id="CLAW1", worker_mode=True,
                     supported_models=["m"], tokens_throughput=100.0)
        r.touch()
        await pm.add_or_update_peer(r)
        assert pm.find_best_worker("m") is not None
        for _ in range(cfg.intervals.max_failed_attempts - 1):
            pm.record_request_failure("CLAW1")
        assert pm.find_best_worker("m") is not None  # below threshold
        pm.record_request_failure("CLAW1")
        assert pm.find_best_worker("m") is None
        assert pm.find_best_worker("m", exclude={"CLAW1"}) is None
    run(go())


def test_health_check_linear_backoff(cfg):
    """Failed health checks push the next check out linearly
    (failed_attempts * backoff_base — reference manager.go:544-548)."""
    async def go():
        disc = Discovery([], _TEST_ID)
        pm = PeerManager(disc, cfg.intervals, self_id="CLAME")
        r = Resource(peer_id="CLAW1", worker_mode=True,
                     supported_models=["m"], tokens_throughput=10.0,
                     addrs=["127.0.0.1:9"])  # refuses connections
        r.touch()
        await pm.add_or_update_peer(r)
        pi = pm.peers["CLAW1"]
        t0 = time.time()
        # emulate two failing health passes (the loop body's except path)
        for expect_fails in (1, 2):
            try:
                await disc.request_metadata(pi.resource.addrs)
            except Exception:
                pi.failed_attempts += 1
                pi.next_health_check = (time.time() +
                                        pi.failed_attempts *
                                        cfg.intervals.backoff_base)
            assert pi.failed_attempts == expect_fails
        # backoff grows with the failure count
        assert pi.next_health_check >= t0 + 2 * cfg.intervals.backoff_base
    run(go())


def test_rendezvous_rejects_malformed(cfg):
    """The rendezvous server answers malformed ops with an error instead of
    dying (and keeps serving)."""
    async def go():
        srv = DHTServer(cfg, "CLADHT")
        port = await srv.start("127.0.0.1", 0)
        cl = RendezvousClient(f"127.0.0.1:{port}", _TEST_ID)
        try:
            r = await cl.call({"op": "bogus"})
            assert r.get("ok") is False
            r = await cl.call({"op": "provide"})  # missing fields
            assert r.get("ok") is False
            assert await cl.ping()  # still alive
        finally:
            await cl.close()
            await srv.stop()
    run(go())


def test_nat_reachability_classification(cfg):
    """Reachability classification (reference NAT stats parity,
    pkg/dht/dht.go:279-309): the rendezvous node classifies providers by
    comparing their observed source address with their advertised addrs,
    and peers can self-classify via the observed_addr ping echo."""
    async def go():
        srv = DHTServer(cfg, "CLADHT")
        port = await srv.start("127.0.0.1", 0)
        cli = RendezvousClient(f"127.0.0.1:{port}", _TEST_ID)
        try:
            obs = await cli.observed_addr()
            assert obs and obs.startswith("127.0.0.1:")
            await cli.provide(_TEST_ID.peer_id, ["127.0.0.1:5001"])
            st = (await cli.stats())
            assert st["nat"]["loopback"] == 1  # loopback test mesh
            # server-side classifier unit cases
            assert DHTServer._classify("10.0.0.5:44", ["10.0.0.5:9"]) == \
                "direct"
            assert DHTServer._classify("198.51.100.7:44",
                                       ["10.0.0.5:9"]) == "translated"
        finally:
            await cli.close()
            await srv.stop()
    run(go())


def test_discovery_learned_servers(cfg):
    """Discovery.add_server: dedupes against bootstrap + learned addrs and
    caps the client list (embedded per-peer rendezvous servers)."""
    ident = identity_from_seed(b"\x07" * 32)
    disco = Discovery(["127.0.0.1:9000"], ident)
    assert disco.add_server("127.0.0.1:9100")          # learned
    assert not disco.add_server("127.0.0.1:9100")      # duplicate
    assert not disco.add_server("127.0.0.1:9000")      # bootstrap dup
    assert not disco.add_server("")                    # empty
    n0 = len(disco.clients)
    for i in range(Discovery.MAX_SERVERS + 4):
        disco.add_server(f"127.0.0.1:{9200 + i}")
    assert len(disco.clients) == Discovery.MAX_SERVERS
    assert n0 <= Discovery.MAX_SERVERS
