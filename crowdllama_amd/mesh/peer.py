"""Unified peer — one struct serves both worker and consumer roles, flipped
by worker_mode (reference parity: pkg/peer/peer.go:42-525).

Serves the inference protocol (length-prefixed protobuf BaseMessage,
5 s read deadline, errors stringified into GenerateResponse — peer.go:190-281)
and the metadata protocol (Resource JSON then close, EOF-delimited —
peer.go:284-316); runs the advertise loop (Provide every 1 s, peer.go:450-504)
and metadata refresh (peer.go:361-389). Worker capabilities are real HIP
device properties + measured throughput, not the reference's hardcoded
"RTX 4090"/150 tok/s (peer.go:319-343)."""

from __future__ import annotations

import asyncio
import time

from ..config import Config
from ..engine.api import EngineBase, RollingRate
from ..keys import load_identity
from ..logutil import new_app_logger
from ..version import __version__, commit_hash
from . import pb
from .discovery import Discovery
from .peermanager import PeerManager
from .resource import Resource
from .wire import (NAMESPACE, PROTO_INFERENCE, PROTO_METADATA,
                   PROTO_RENDEZVOUS, secure_accept)


class Peer:
    def __init__(self, cfg: Config, worker_mode: bool,
                 engines: dict[str, EngineBase] | None = None):
        self.cfg = cfg
        self.worker_mode = worker_mode
        component = "worker" if worker_mode else "consumer"
        self.identity = load_identity(component, cfg.key_path)
        self.peer_id = self.identity.peer_id
        self.log = new_app_logger(f"peer.{component}", cfg.verbose)
        self.engines: dict[str, EngineBase] = engines or {}
        self.discovery = Discovery(
            cfg.bootstrap_peers, self.identity,
            metadata_timeout=cfg.intervals.metadata_timeout,
            metadata_max_age=cfg.intervals.metadata_max_age,
            log=self.log)
        self.peer_manager = PeerManager(self.discovery, cfg.intervals,
                                        log=self.log, self_id=self.peer_id)
        # gossip fallback: when every bootstrap node is unreachable,
        # discovery re-resolves providers through known healthy peers
        self.discovery.fallback_addrs = self._gossip_addrs
        self.rate = RollingRate()
        self._active_requests = 0
        self.resource = Resource(peer_id=self.peer_id,
                                 worker_mode=worker_mode,
                                 version=f"{__version__}+{commit_hash()}")
        self._server: asyncio.base_events.Server | None = None
        self._dht = None              # embedded rendezvous server
        self._conn_writers: set = set()
        self._tasks: list[asyncio.Task] = []
        self.port: int | None = None
        self.requests_served = 0

    # ----------------------------------------------------------- lifecycle

    async def start(self) -> None:
        self._server = await asyncio.start_server(
            self._on_conn, self.cfg.listen_host, self.cfg.listen_port)
        self.port = self._server.sockets[0].getsockname()[1]
        if self.cfg.peer_dht:
            # every peer runs an embedded rendezvous server (reference:
            # libp2p DHT ModeServer on every peer, pkg/dht/dht.go:106-112)
            # and advertises it in its signed record; consumers add it as
            # a rendezvous target, so the mesh keeps N live servers after
            # the bootstrap node dies instead of relying only on gossip
            from .dhtnode import DHTServer
            self._dht = DHTServer(self.cfg, identity=self.identity)
            dht_port = await self._dht.start(self.cfg.listen_host, 0)
            host = "127.0.0.1" if self.cfg.listen_host in ("0.0.0.0", "::")                 else self.cfg.listen_host
            self.resource.dht_addr = f"{host}:{dht_port}"
            # self-register: the advertise loop then keeps this peer's
            # record live on its OWN server, so a consumer that learned
            # it can resolve providers there after the bootstrap dies
            self.discovery.add_server(self.resource.dht_addr)
        self.update_metadata()
        await self.peer_manager.start()
        self._tasks = [
            asyncio.create_task(self._advertise_loop()),
            asyncio.create_task(self._metadata_update_loop()),
        ]
        self.log.info("peer %s (%s) listening on :%d, bootstrap=%s",
                      self.peer_id, "worker" if self.worker_mode else
                      "consumer", self.port, self.cfg.bootstrap_peers)

    async def stop(self) -> None:
        for t in self._tasks:
            t.cancel()
        for t in self._tasks:
            try:
                await t
            except (asyncio.CancelledError, Exception):
                pass
        self._tasks.clear()
        await self.peer_manager.stop()
        await self.discovery.close()
        if self._dht is not None:
            await self._dht.stop()
            self._dht = None
        if self._server:
            self._server.close()
        for w in list(self._conn_writers):   # see DHTServer.stop: close
            try:                             # BEFORE wait_closed or 3.10
                w.close()                    # deadlocks on live handlers
            except Exception:
                pass
        self._conn_writers.clear()
        if self._server:
            try:
                await asyncio.wait_for(self._server.wait_closed(), 5)
            except asyncio.TimeoutError:
                pass
        for e in self.engines.values():
            await e.close()

    @property
    def addrs(self) -> list[str]:
        host = "127.0.0.1" if self.cfg.listen_host in ("0.0.0.0", "::") \
            else self.cfg.listen_host
        return [f"{host}:{self.port}"]

    # ------------------------------------------------------------ metadata

    def update_metadata(self) -> None:
        """Fill the advertised Resource from live engine state
        (reference UpdateMetadata, peer.go:319-343 — but measured)."""
        r = self.resource
        if self.worker_mode and self.engines:
            r.supported_models = sorted(self.engines.keys())
            any_engine = next(iter(self.engines.values()))
            r.gpu_model = any_engine.gpu_model()
            r.vram_gb = any_engine.vram_gb()
            measured = self.rate.rate()
            r.tokens_throughput = measured if measured > 0 else \
                max(any_engine.throughput(), 1.0)
            r.load = min(1.0, self._active_requests / 4.0)
        else:
            r.supported_models = []
            r.tokens_throughput = 0.0
            r.load = 0.0
        r.addrs = self.addrs
        r.touch()
        # signed record: receivers verify the signature and that peer_id
        # is the hash of this public key (mesh/resource.py verify())
        r.sign(self.identity.seed, self.identity.pub)

    async def _metadata_update_loop(self) -> None:
        while True:
            await asyncio.sleep(self.cfg.intervals.metadata_update)
            self.update_metadata()

    async def _advertise_loop(self) -> None:
        # reference: Provide namespace CID every 1 s (peer.go:450-504); also
        # per-model namespaces (AdvertiseModel, discovery.go:144-166).
        while True:
            try:
                await self.discovery.advertise(self.peer_id, self.addrs,
                                               NAMESPACE)
                if self.worker_mode:
                    for model in self.engines:
                        await self.discovery.advertise(
                            self.peer_id, self.addrs,
                            f"{NAMESPACE}/model/{model}")
            except Exception as e:  # noqa: BLE001
                self.log.debug("advertise failed: %s", e)
            await asyncio.sleep(self.cfg.intervals.advertise)

    # -------------------------------------------------------------- serving

    async def _on_conn(self, reader: asyncio.StreamReader,
                       writer: asyncio.StreamWriter) -> None:
        self._conn_writers.add(writer)
        try:
            # every mesh stream is encrypted + mutually authenticated
            # (wire.py secure_accept; the reference gets this from libp2p
            # noise/TLS — discovery.go:48-84)
            ss, proto = await secure_accept(reader, writer, self.identity)
            if proto == PROTO_METADATA:
                self.update_metadata()
                await ss.write_frame(self.resource.to_json().encode("utf-8"))
                return
            if proto == PROTO_RENDEZVOUS:
                # read-only rendezvous served from this peer's OWN registry
                # (gossip fallback): the mesh survives the bootstrap node
                # dying — consumers re-resolve providers from any live peer
                # (reference parity: every libp2p peer runs the DHT in
                # ModeServer, pkg/dht/dht.go:106-112 / discovery.go:92-141)
                await self._handle_rendezvous(ss)
                return
            if proto == PROTO_INFERENCE:
                # serve multiple sequential requests per connection
                # (keep-alive extension: the gateway pools authenticated
                # streams so the handshake amortizes across requests; the
                # reference opens one libp2p stream per request)
                await self._handle_inference(ss)
                while True:
                    await self._handle_inference(ss, idle_timeout=120.0)
            self.log.warning("unknown protocol %r", proto)
        except Exception as e:  # noqa: BLE001
            self.log.debug("conn error: %s", e)
        finally:
            self._conn_writers.discard(writer)
            try:
                writer.close()
            except Exception:
                pass

    async def _handle_rendezvous(self, ss) -> None:
        import asyncio as _aio
        while True:
            try:
                msg = await ss.read_json(timeout=60.0)
            except (_aio.IncompleteReadError, _aio.TimeoutError,
                    ConnectionError, ValueError):
                return
            op = msg.get("op")
            if op == "ping":
                await ss.write_json({"ok": True, "peer_id": self.peer_id})
            elif op == "find_providers":
                ns = msg.get("ns", "")
                limit = int(msg.get("limit", 10))
                out = []
                # this peer itself is a provider of its namespaces
                if self.worker_mode and (
                        ns == NAMESPACE or any(
                            ns == f"{NAMESPACE}/model/{m}"
                            for m in self.engines)):
                    out.append({"peer_id": self.peer_id,
                                "addrs": self.addrs})
                for pid, pi in self.peer_manager.peers.items():
                    if len(out) >= limit:
                        break
                    r = pi.resource
                    if not pi.is_healthy or not r.worker_mode:
                        continue
                    if ns != NAMESPACE and not any(
                            ns == f"{NAMESPACE}/model/{m}"
                            for m in r.supported_models):
                        continue
                    out.append({"peer_id": pid, "addrs": r.addrs})
                await ss.write_json({"ok": True, "providers": out[:limit]})
            elif op == "find_peer":
                pid = msg.get("peer_id", "")
                if pid == self.peer_id:
                    await ss.write_json({"ok": True, "addrs": self.addrs})
                    continue
                pi = self.peer_manager.peers.get(pid)
                if pi is None:
                    await ss.write_json({"ok": False,
                                         "error": "peer not found"})
                else:
                    await ss.write_json({"ok": True,
                                         "addrs": pi.resource.addrs})
            else:
                # provide/remove are bootstrap-node ops; gossip is read-only
                await ss.write_json({"ok": False,
                                     "error": f"unsupported op {op!r}"})

    async def _handle_inference(self, ss, idle_timeout: float = 5.0) -> None:
        # first-request read deadline parity: 5 s (peer.go:259-271);
        # keep-alive waits longer between pooled requests
        frame = await ss.read_frame(timeout=idle_timeout)
        msg = pb.BaseMessage.decode(frame)
        if not self.worker_mode:
            resp = pb.response_message("", "Error: peer is not a worker",
                                       self.peer_id, done_reason="error")
            await ss.write_frame(resp.encode())
            return
        req = msg.generate_request
        if req is None:
            resp = pb.response_message("", "Error: no request in message",
                                       self.peer_id, done_reason="error")
            await ss.write_frame(resp.encode())
            return
        t0 = time.monotonic_ns()
        self._active_requests += 1
        try:
            engine = self.engines.get(req.model)
            if engine is None:
                raise KeyError(f"model {req.model!r} not served here "
                               f"(have {sorted(self.engines)})")
            if req.stream:
                # streamed chunks: one done=False frame per text delta, then
                # a final done=True frame (extension over the reference,
                # which reserves `stream` but never streams — SURVEY.md §2.2)
                total_tokens = 0
                async for chunk in engine.generate_stream(req.prompt):
                    total_tokens = chunk.tokens_generated
                    final = bool(chunk.done_reason)
                    resp = pb.response_message(
                        req.model, chunk.text, self.peer_id,
                        done_reason=chunk.done_reason, done=final,
                        total_duration_ns=(time.monotonic_ns() - t0
                                           if final else 0),
                        eval_count=chunk.tokens_generated)
                    await ss.write_frame(resp.encode())
                self.rate.add(total_tokens)
                self.requests_served += 1
                return
            result = await engine.generate(req.prompt)
            self.rate.add(result.tokens_generated)
            self.requests_served += 1
            resp = pb.response_message(
                req.model, result.text, self.peer_id,
                done_reason=result.done_reason,
                total_duration_ns=time.monotonic_ns() - t0,
                eval_count=result.tokens_generated)
        except Exception as e:  # noqa: BLE001 — reference stringifies errors
            resp = pb.response_message(req.model, f"Error: {e}", self.peer_id,
                                       done_reason="error",
                                       total_duration_ns=time.monotonic_ns() - t0)
        finally:
            self._active_requests -= 1
        await ss.write_frame(resp.encode())

    def _gossip_addrs(self) -> list[str]:
        out = []
        for pi in self.peer_manager.peers.values():
            if pi.is_healthy and pi.resource.addrs:
                out.extend(pi.resource.addrs)
        return out

    async def reachability(self) -> str:
        """Classify this peer's reachability from a bootstrap node's view
        (reference NAT classification, pkg/dht/dht.go:346-395): "direct"
        when the observed address matches an advertised one, "translated"
        behind a NAT/proxy, "loopback" for local meshes, "unknown" when no
        bootstrap answers. Hole punching / UPnP port mapping are a
        documented descope (docs/PARITY.md): deploy workers with a
        reachable address or a TCP reverse proxy."""
        for c in self.discovery.clients:
            obs = await c.observed_addr()
            if not obs:
                continue
            host = obs.rsplit(":", 1)[0]
            if host in ("127.0.0.1", "::1", "localhost"):
                return "loopback"
            adv = {a.rsplit(":", 1)[0] for a in self.addrs}
            return "direct" if host in adv else "translated"
        return "unknown"

    def is_dht_connected(self) -> bool:
        """True while rendezvous round trips are succeeding (reference
        IsDHTConnected checks the routing table, peer.go:513-525; here the
        equivalent liveness signal is a recent successful provide/find)."""
        if not self.cfg.bootstrap_peers:
            return False
        window = 3.0 * max(self.cfg.intervals.advertise, 1.0)
        return (time.monotonic() - self.discovery.last_success) < window
