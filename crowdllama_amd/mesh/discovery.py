"""Discovery client (reference parity: internal/discovery/discovery.go).

Wraps rendezvous RPCs (provide / find_providers / find_peer — the analog of
DHT Provide/FindProvidersAsync/FindPeer) and the peer metadata protocol
(RequestPeerMetadata, discovery.go:186-275: 5 s deadline, EOF-delimited
Resource JSON)."""

from __future__ import annotations

import asyncio
import logging
import time

from .resource import Resource
from .wire import (NAMESPACE, PROTO_METADATA, PROTO_RENDEZVOUS,
                   SecureStream, secure_open)


def parse_addr(addr: str) -> tuple[str, int]:
    host, _, port = addr.rpartition(":")
    return host or "127.0.0.1", int(port)


class RendezvousClient:
    """Persistent, encrypted connection to one bootstrap node."""

    def __init__(self, addr: str, identity, timeout: float = 5.0):
        self.addr = addr
        self.identity = identity
        self.timeout = timeout
        self._ss: SecureStream | None = None
        self._lock = asyncio.Lock()

    async def _ensure(self):
        if self._ss is None or self._ss.is_closing():
            host, port = parse_addr(self.addr)
            self._ss = await secure_open(host, port, PROTO_RENDEZVOUS,
                                         self.identity, self.timeout)

    async def call(self, msg: dict) -> dict:
        async with self._lock:
            try:
                await self._ensure()
                await self._ss.write_json(msg)
                return await self._ss.read_json(self.timeout)
            except Exception:
                # one reconnect attempt per call
                await self.close()
                await self._ensure()
                await self._ss.write_json(msg)
                return await self._ss.read_json(self.timeout)

    async def close(self):
        if self._ss is not None:
            self._ss.close()
        self._ss = None

    async def ping(self) -> bool:
        try:
            r = await self.call({"op": "ping"})
            return bool(r.get("ok"))
        except Exception:
            return False

    async def observed_addr(self) -> str | None:
        """This client's address as seen by the bootstrap node (the
        analog of libp2p identify's observed address)."""
        try:
            r = await self.call({"op": "ping"})
            return r.get("observed_addr") if r.get("ok") else None
        except Exception:
            return None

    async def provide(self, peer_id: str, addrs: list[str],
                      ns: str = NAMESPACE) -> bool:
        r = await self.call({"op": "provide", "ns": ns, "peer_id": peer_id,
                             "addrs": addrs})
        return bool(r.get("ok"))

    async def find_providers(self, ns: str = NAMESPACE,
                             limit: int = 10) -> list[dict]:
        r = await self.call({"op": "find_providers", "ns": ns, "limit": limit})
        return r.get("providers", []) if r.get("ok") else []

    async def stats(self) -> dict | None:
        "Rendezvous-server statistics (dht stats op)."
        r = await self.call({"op": "stats"})
        return r.get("stats") if r.get("ok") else None

    async def find_peer(self, peer_id: str) -> list[str] | None:
        r = await self.call({"op": "find_peer", "peer_id": peer_id})
        return r.get("addrs") if r.get("ok") else None


class Discovery:
    """Multi-bootstrap discovery + metadata fetch (reference Discovery)."""

    def __init__(self, bootstrap_addrs: list[str], identity,
                 metadata_timeout: float = 5.0,
                 metadata_max_age: float = 3600.0,
                 log: logging.Logger | None = None):
        self.identity = identity
        self.clients = [RendezvousClient(a, identity,
                                         timeout=metadata_timeout)
                        for a in bootstrap_addrs]
        # gossip fallback (de-SPOF): a callable returning live peer addrs
        # to query when every bootstrap node is down (Peer wires this to
        # its peermanager; reference parity: every libp2p peer is a DHT
        # server, pkg/dht/dht.go:106-112)
        self.fallback_addrs = None
        self.metadata_timeout = metadata_timeout
        self.metadata_max_age = metadata_max_age
        self.log = log or logging.getLogger("discovery")
        # monotonic time of the last successful rendezvous round trip;
        # Peer.is_dht_connected derives liveness from this (the reference
        # checks the DHT routing table instead, peer.go:513-525)
        self.last_success = 0.0

    MAX_SERVERS = 8

    def add_server(self, addr: str, peer_id: str = "") -> bool:
        """Add a learned rendezvous server (a peer's embedded DHT).
        Every peer runs one (reference: libp2p ModeServer on every peer),
        so after bootstrap the mesh has N rendezvous targets, not one."""
        if not addr or len(self.clients) >= self.MAX_SERVERS:
            return False
        if any(c.addr == addr for c in self.clients):
            return False
        self.clients.append(RendezvousClient(addr, self.identity,
                                             timeout=self.metadata_timeout))
        self.log.info("learned rendezvous server %s", addr)
        return True

    async def close(self):
        for c in self.clients:
            await c.close()

    async def bootstrap_ok(self) -> bool:
        for c in self.clients:
            if await c.ping():
                return True
        return False

    async def advertise(self, peer_id: str, addrs: list[str],
                        ns: str = NAMESPACE) -> None:
        for c in self.clients:
            try:
                await c.provide(peer_id, addrs, ns)
                self.last_success = time.monotonic()
            except Exception as e:
                self.log.debug("advertise to %s failed: %s", c.addr, e)

    async def find_providers(self, ns: str = NAMESPACE,
                             limit: int = 10) -> list[dict]:
        seen = {}
        ok = False
        for c in self.clients:
            try:
                for p in await c.find_providers(ns, limit):
                    seen[p["peer_id"]] = p
                ok = True
                self.last_success = time.monotonic()
            except Exception as e:
                self.log.debug("find_providers on %s failed: %s", c.addr, e)
        if not ok and self.fallback_addrs is not None:
            # bootstrap outage: ask known live peers (gossip, read-only)
            for addr in self.fallback_addrs()[:5]:
                try:
                    cli = RendezvousClient(addr, self.identity,
                                           timeout=self.metadata_timeout)
                    try:
                        for p in await cli.find_providers(ns, limit):
                            seen[p["peer_id"]] = p
                    finally:
                        await cli.close()
                    self.last_success = time.monotonic()
                except Exception as e:  # noqa: BLE001
                    self.log.debug("gossip find on %s failed: %s", addr, e)
        return list(seen.values())[:limit]

    async def find_peer_addrs(self, peer_id: str) -> list[str] | None:
        for c in self.clients:
            try:
                addrs = await c.find_peer(peer_id)
                if addrs:
                    return addrs
            except Exception:
                continue
        return None

    async def request_metadata(self, addrs: list[str],
                               expected_peer_id: str | None = None
                               ) -> Resource:
        """Open a metadata stream and read one signed Resource frame
        (reference discovery.go:186-275 reads EOF-delimited JSON over a
        libp2p-secured stream; here the channel handshake authenticates
        the peer AND the record carries its own signature)."""
        last_err: Exception | None = None
        for addr in addrs:
            host, port = parse_addr(addr)
            try:
                ss = await secure_open(host, port, PROTO_METADATA,
                                       self.identity, self.metadata_timeout,
                                       expected_peer_id=expected_peer_id)
                try:
                    data = await ss.read_frame(self.metadata_timeout)
                    res = Resource.from_json(data)
                    if not res.verify():
                        raise ValueError("resource signature invalid")
                    if res.peer_id != ss.peer_id:
                        raise ValueError("resource peer_id mismatch")
                    if res.age() > self.metadata_max_age:
                        raise ValueError("metadata too stale")
                    return res
                finally:
                    ss.close()
            except Exception as e:  # noqa: BLE001
                last_err = e
        raise last_err or ConnectionError("no addresses")

    async def discover_peers(self, limit: int = 10,
                             ns: str = NAMESPACE) -> list[Resource]:
        """FindProviders + per-provider metadata fetch
        (reference DiscoverPeers, discovery.go:332-366)."""
        out = []
        for p in await self.find_providers(ns, limit):
            try:
                res = await self.request_metadata(
                    p.get("addrs", []), expected_peer_id=p["peer_id"])
                if not res.addrs:
                    res.addrs = p.get("addrs", [])
                out.append(res)
            except Exception as e:  # noqa: BLE001
                self.log.debug("metadata fetch from %s failed: %s",
                               p.get("peer_id"), e)
        return out
