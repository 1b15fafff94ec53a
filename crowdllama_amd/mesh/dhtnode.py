"""Rendezvous/bootstrap node — the MI355X mesh analog of the reference's
Kademlia DHT server (pkg/dht/dht.go:25-437).

The reference uses libp2p Kademlia purely as a rendezvous: workers
`Provide` a namespace CID every second and consumers `FindProvidersAsync`
it. This node implements those semantics directly over the
mesh wire protocol: provider records with TTL expiry, peer-address lookup,
connection stats logging, and eager eviction on disconnect
(dht.go:370-383).
"""

from __future__ import annotations

import asyncio
import time
from dataclasses import dataclass, field

from ..config import Config
from ..keys import load_identity
from ..logutil import new_app_logger
from .wire import PROTO_RENDEZVOUS, secure_accept


@dataclass
class ProviderRecord:
    peer_id: str
    addrs: list[str]
    last_provide: float = field(default_factory=time.time)


class DHTServer:
    """Standalone bootstrap/rendezvous node (reference pkg/dht Server)."""

    PROVIDER_TTL = 30.0  # records expire without re-provide (ref: 1 s loop)

    def __init__(self, cfg: Config, peer_id: str = "", identity=None):
        self.cfg = cfg
        # real cryptographic identity from the keyfile (the legacy second
        # argument is kept for compatibility but the advertised id is
        # always derived from the key — ids are now verifiable hashes of
        # ed25519 public keys). An embedded per-peer
        # server passes the peer's own identity instead.
        self.identity = identity or load_identity("dht", cfg.key_path)
        self.peer_id = self.identity.peer_id
        self.log = new_app_logger("dht", cfg.verbose)
        self.providers: dict[str, dict[str, ProviderRecord]] = {}  # ns -> id -> rec
        self.peer_addrs: dict[str, list[str]] = {}
        # reachability classification per peer (reference NAT stats,
        # pkg/dht/dht.go:279-309,346-395): the server compares each
        # provider's OBSERVED source address with its advertised addrs.
        # "direct" = advertised host matches what we see; "translated" =
        # mismatch (NAT/proxy in the path); "loopback" = local testing.
        self.peer_reachability: dict[str, str] = {}
        self.conn_count = 0
        self.total_conns = 0
        self._server: asyncio.base_events.Server | None = None
        self._tasks: list[asyncio.Task] = []
        self._conn_writers: set = set()
        self.port: int | None = None

    async def start(self, host: str | None = None, port: int | None = None):
        host = host or self.cfg.listen_host
        port = self.cfg.dht_port if port is None else port
        self._server = await asyncio.start_server(self._on_conn, host, port)
        self.port = self._server.sockets[0].getsockname()[1]
        self._tasks.append(asyncio.create_task(self._stats_loop()))
        self._tasks.append(asyncio.create_task(self._expiry_loop()))
        self.log.info("DHT server %s listening on %s:%d",
                      self.peer_id, host, self.port)
        return self.port

    async def stop(self):
        for t in self._tasks:
            t.cancel()
        for t in self._tasks:
            try:
                await t
            except (asyncio.CancelledError, Exception):
                pass
        self._tasks.clear()
        if self._server:
            self._server.close()
        # asyncio's Server.close() only stops ACCEPTING; live connection
        # handlers would keep answering old clients from this (stopped)
        # node's state — close them so clients fail over/reconnect. This
        # must happen BEFORE wait_closed(): on 3.10 wait_closed blocks
        # until the last live handler detaches, so waiting first
        # deadlocks against a peer holding a persistent connection.
        for w in list(self._conn_writers):
            try:
                w.close()
            except Exception:
                pass
        self._conn_writers.clear()
        if self._server:
            try:
                await asyncio.wait_for(self._server.wait_closed(), 5)
            except asyncio.TimeoutError:
                pass
            self._server = None

    # ------------------------------------------------------------ serving

    async def _on_conn(self, reader: asyncio.StreamReader,
                       writer: asyncio.StreamWriter):
        self.conn_count += 1
        self.total_conns += 1
        self._conn_writers.add(writer)
        try:
            # authenticated, encrypted channel: the caller's peer_id is
            # cryptographically verified before any op is served
            ss, proto = await secure_accept(reader, writer, self.identity)
            if proto != PROTO_RENDEZVOUS:
                self.log.warning("unknown protocol %r", proto)
                return
            peername = writer.get_extra_info("peername") or ("?", 0)
            observed = f"{peername[0]}:{peername[1]}"
            while True:
                try:
                    msg = await ss.read_json(timeout=60.0)
                except (asyncio.IncompleteReadError, asyncio.TimeoutError,
                        ConnectionError, ValueError):
                    return
                resp = self._handle(msg, ss.peer_id, observed)
                await ss.write_json(resp)
        except Exception as e:  # noqa: BLE001 — per-conn isolation
            self.log.debug("conn error: %s", e)
        finally:
            self.conn_count -= 1
            self._conn_writers.discard(writer)
            writer.close()

    def _handle(self, msg: dict, caller_id: str = "",
                observed: str = "") -> dict:
        op = msg.get("op")
        if op == "ping":
            # observed_addr lets the caller classify its own reachability
            # (the analog of libp2p's identify/observed-address)
            return {"ok": True, "peer_id": self.peer_id,
                    "observed_addr": observed}
        if op == "provide":
            ns = msg.get("ns", "")
            pid = msg.get("peer_id", "")
            addrs = list(msg.get("addrs", []))
            if not ns or not pid:
                return {"ok": False, "error": "missing ns/peer_id"}
            if caller_id and pid != caller_id:
                # a peer may only advertise ITS OWN (proven) identity
                return {"ok": False,
                        "error": f"peer_id {pid} does not match "
                                 f"authenticated identity"}
            self.providers.setdefault(ns, {})[pid] = ProviderRecord(
                pid, addrs)
            self.peer_addrs[pid] = addrs
            self.peer_reachability[pid] = self._classify(observed, addrs)
            return {"ok": True}
        if op == "find_providers":
            ns = msg.get("ns", "")
            limit = int(msg.get("limit", 10))  # ref: FindProvidersAsync(cid,10)
            out = []
            now = time.time()
            for rec in list(self.providers.get(ns, {}).values()):
                if now - rec.last_provide > self.PROVIDER_TTL:
                    continue
                out.append({"peer_id": rec.peer_id, "addrs": rec.addrs})
                if len(out) >= limit:
                    break
            return {"ok": True, "providers": out}
        if op == "find_peer":
            pid = msg.get("peer_id", "")
            addrs = self.peer_addrs.get(pid)
            if addrs is None:
                return {"ok": False, "error": "peer not found"}
            return {"ok": True, "addrs": addrs}
        if op == "remove":
            pid = msg.get("peer_id", "")
            if caller_id and pid != caller_id:
                return {"ok": False, "error": "cannot remove another peer"}
            for ns in self.providers.values():
                ns.pop(pid, None)
            self.peer_addrs.pop(pid, None)
            return {"ok": True}
        if op == "stats":
            return {"ok": True, "stats": self.stats()}
        return {"ok": False, "error": f"unknown op {op!r}"}

    @staticmethod
    def _classify(observed: str, advertised: list[str]) -> str:
        host = observed.rsplit(":", 1)[0]
        if host in ("127.0.0.1", "::1", "localhost"):
            return "loopback"
        adv_hosts = {a.rsplit(":", 1)[0] for a in advertised}
        return "direct" if host in adv_hosts else "translated"

    def nat_stats(self) -> dict:
        out = {"direct": 0, "translated": 0, "loopback": 0}
        for v in self.peer_reachability.values():
            out[v] = out.get(v, 0) + 1
        return out

    def stats(self) -> dict:
        nprov = sum(len(v) for v in self.providers.values())
        return {
            "peer_id": self.peer_id,
            "known_peers": len(self.peer_addrs),
            "providers": nprov,
            "namespaces": len(self.providers),
            "active_conns": self.conn_count,
            "total_conns": self.total_conns,
            "nat": self.nat_stats(),
        }

    # ----------------------------------------------------------- bg loops

    async def _stats_loop(self):
        # reference: NAT/peer stats every 30 s / 15 s (dht.go:279-321)
        while True:
            await asyncio.sleep(self.cfg.intervals.nat_log)
            s = self.stats()
            self.log.info("stats: peers=%d providers=%d conns=%d/%d "
                          "nat=%s", s["known_peers"], s["providers"],
                          s["active_conns"], s["total_conns"], s["nat"])

    async def _expiry_loop(self):
        while True:
            await asyncio.sleep(self.PROVIDER_TTL / 2)
            now = time.time()
            for ns, recs in self.providers.items():
                dead = [pid for pid, r in recs.items()
                        if now - r.last_provide > self.PROVIDER_TTL]
                for pid in dead:
                    del recs[pid]
                    self.log.debug("expired provider %s in %s", pid, ns)
