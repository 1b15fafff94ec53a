"""Peer membership + scheduling (reference parity: pkg/peermanager/manager.go).

Semantics carried over exactly: thread-safe registry
with recently-removed tombstones, four background loops (discovery, metadata
touch, health check, cleanup), health = reachability + metadata RTT with
per-peer linear backoff `failed_attempts * backoff_base`, stale eviction
after 1 min, 10 min tombstones, and the scheduler maximizing
`tokens_throughput / (1 + load)` (manager.go:338-387)."""

from __future__ import annotations

import asyncio
import logging
import time
from dataclasses import dataclass, field

from ..config import Intervals
from .discovery import Discovery
from .resource import Resource


@dataclass
class PeerInfo:
    resource: Resource
    last_seen: float = field(default_factory=time.time)
    failed_attempts: int = 0
    next_health_check: float = 0.0
    is_healthy: bool = True


class PeerManager:
    def __init__(self, discovery: Discovery, intervals: Intervals,
                 log: logging.Logger | None = None,
                 self_id: str = ""):
        self.discovery = discovery
        self.iv = intervals
        self.log = log or logging.getLogger("peermanager")
        self.self_id = self_id
        self.peers: dict[str, PeerInfo] = {}
        self.recently_removed: dict[str, float] = {}  # tombstones
        self._tasks: list[asyncio.Task] = []
        self._lock = asyncio.Lock()

    # ------------------------------------------------------------- registry

    async def add_or_update_peer(self, res: Resource) -> None:
        async with self._lock:
            if res.peer_id == self.self_id:
                return
            ts = self.recently_removed.get(res.peer_id)
            if ts is not None:
                if time.time() - ts < self.iv.tombstone:
                    return  # tombstoned: prevent re-add flapping
                del self.recently_removed[res.peer_id]
            pi = self.peers.get(res.peer_id)
            if pi is None:
                self.peers[res.peer_id] = PeerInfo(resource=res)
                self.log.info("peer added: %s (models=%s, %.0f tok/s)",
                              res.peer_id, res.supported_models,
                              res.tokens_throughput)
            else:
                pi.resource = res
                pi.last_seen = time.time()
            if res.dht_addr:
                # the record is signature-verified upstream, so the
                # embedded-server address is as trustworthy as the rest
                self.discovery.add_server(res.dht_addr, res.peer_id)

    async def remove_peer(self, peer_id: str, tombstone: bool = True) -> None:
        async with self._lock:
            if self.peers.pop(peer_id, None) is not None:
                self.log.info("peer removed: %s", peer_id)
            if tombstone:
                self.recently_removed[peer_id] = time.time()

    def is_peer_unhealthy(self, peer_id: str) -> bool:
        pi = self.peers.get(peer_id)
        return pi is None or not pi.is_healthy

    def get_healthy_peers(self) -> list[Resource]:
        return [pi.resource for pi in self.peers.values() if pi.is_healthy]

    # ------------------------------------------------------------ scheduler

    def find_best_worker(self, model: str,
                         exclude: set[str] | None = None,
                         extra_load: dict[str, float] | None = None
                         ) -> Resource | None:
        """Max of tokens_throughput/(1+load) over healthy workers supporting
        the model (reference manager.go:338-387). Extensions over the
        reference: `exclude` skips peers that already failed this request
        (gateway failover), and `extra_load` adds the caller's own in-flight
        request count per peer — advertised load is seconds stale, so
        without it a burst of concurrent requests all tie-break onto the
        same worker."""
        best, best_score = None, -1.0
        for pi in self.peers.values():
            r = pi.resource
            if exclude and r.peer_id in exclude:
                continue
            if not pi.is_healthy or not r.worker_mode:
                continue
            if model and model not in r.supported_models:
                continue
            load = max(0.0, r.load)
            if extra_load:
                load += max(0.0, extra_load.get(r.peer_id, 0.0))
            score = r.tokens_throughput / (1.0 + load)
            if score > best_score:
                best, best_score = r, score
        return best

    def record_request_failure(self, peer_id: str) -> None:
        """Count an inference-path failure against the peer's health
        (same counter and threshold the health checker uses,
        manager.go:536-622); a recovered health check resets it."""
        pi = self.peers.get(peer_id)
        if pi is not None:
            pi.failed_attempts += 1
            if pi.failed_attempts >= self.iv.max_failed_attempts:
                pi.is_healthy = False

    def get_peer_statistics(self) -> dict:
        healthy = sum(1 for p in self.peers.values() if p.is_healthy)
        workers = sum(1 for p in self.peers.values()
                      if p.resource.worker_mode)
        return {
            "total_peers": len(self.peers),
            "healthy_peers": healthy,
            "unhealthy_peers": len(self.peers) - healthy,
            "workers": workers,
            "tombstones": len(self.recently_removed),
        }

    # ------------------------------------------------------------ lifecycle

    async def start(self) -> None:
        self._tasks = [
            asyncio.create_task(self._discovery_loop()),
            asyncio.create_task(self._health_loop()),
            asyncio.create_task(self._cleanup_loop()),
        ]

    async def stop(self) -> None:
        for t in self._tasks:
            t.cancel()
        for t in self._tasks:
            try:
                await t
            except (asyncio.CancelledError, Exception):
                pass
        self._tasks.clear()

    async def _discovery_loop(self) -> None:
        while True:
            try:
                for res in await self.discovery.discover_peers():
                    if self.is_peer_unhealthy(res.peer_id) and \
                            res.peer_id in self.peers:
                        continue  # skip known-unhealthy (gateway.go:383-423)
                    await self.add_or_update_peer(res)
            except Exception as e:  # noqa: BLE001
                self.log.debug("discovery failed: %s", e)
            await asyncio.sleep(self.iv.discovery)

    async def _health_loop(self) -> None:
        while True:
            await asyncio.sleep(self.iv.health_check)
            now = time.time()
            for pid, pi in list(self.peers.items()):
                if now < pi.next_health_check:
                    continue  # linear backoff window
                try:
                    # authenticated probe: the metadata fetch fails unless
                    # the responder proves it IS pid (wire.py secure_open)
                    res = await self.discovery.request_metadata(
                        pi.resource.addrs, expected_peer_id=pid)
                    pi.resource = res if res.peer_id else pi.resource
                    pi.last_seen = now
                    pi.failed_attempts = 0
                    pi.is_healthy = True
                    pi.next_health_check = 0.0
                except Exception:
                    pi.failed_attempts += 1
                    # linear backoff (manager.go:544-548)
                    pi.next_health_check = (
                        now + pi.failed_attempts * self.iv.backoff_base)
                    if pi.failed_attempts >= self.iv.max_failed_attempts:
                        pi.is_healthy = False

    async def _cleanup_loop(self) -> None:
        while True:
            await asyncio.sleep(self.iv.cleanup)
            now = time.time()
            for pid, pi in list(self.peers.items()):
                if now - pi.last_seen > self.iv.stale_timeout:
                    await self.remove_peer(pid)
            # expire tombstones (manager.go:264-271)
            for pid, ts in list(self.recently_removed.items()):
                if now - ts > self.iv.tombstone:
                    del self.recently_removed[pid]
