"""Consumer HTTP gateway — Ollama-compatible API (reference parity:
pkg/gateway/gateway.go).

Serves POST /api/chat + GET /api/health on :9001 with request-logging
middleware (gateway.go:87-154); handleChat: validate -> FindBestWorker ->
503 if none -> RequestInference over the mesh inference protocol -> map PB
response to Ollama-style JSON (gateway.go:168-231). Non-streaming, like the
reference (gateway.go:243-293 is strictly one-shot). Background discovery
pulls DHT providers into the peer manager (gateway.go:351-423)."""

from __future__ import annotations

import asyncio
import json
import time

from aiohttp import web

from ..config import Config
from ..logutil import new_app_logger
from . import pb
from .discovery import parse_addr
from .peer import Peer
from .wire import PROTO_INFERENCE, secure_open


class Gateway:
    def __init__(self, peer: Peer, cfg: Config):
        self.peer = peer
        self.cfg = cfg
        self.log = new_app_logger("gateway", cfg.verbose)
        self.app = web.Application(middlewares=[self._log_middleware])
        self.app.router.add_post("/api/chat", self.handle_chat)
        self.app.router.add_post("/api/generate", self.handle_generate)
        self.app.router.add_get("/api/health", self.handle_health)
        self.app.router.add_get("/api/tags", self.handle_tags)
        self._runner: web.AppRunner | None = None
        self.port: int | None = None
        self._tasks: list[asyncio.Task] = []
        # local in-flight requests per worker; advertised load is seconds
        # stale, so the scheduler folds this in (find_best_worker extra_load)
        self._inflight: dict[str, int] = {}
        # pooled authenticated worker connections: the SIGMA handshake
        # (~20 ms of pure-python ed25519/X25519) amortizes across requests
        self._conns: dict[str, list] = {}

    # ----------------------------------------------------------- lifecycle

    async def start(self, port: int | None = None) -> int:
        self._runner = web.AppRunner(self.app, access_log=None)
        await self._runner.setup()
        site = web.TCPSite(self._runner, self.cfg.listen_host,
                           self.cfg.gateway_port if port is None else port)
        await site.start()
        self.port = site._server.sockets[0].getsockname()[1]  # noqa: SLF001
        self._tasks.append(asyncio.create_task(self._discovery_loop()))
        self.log.info("gateway listening on :%d", self.port)
        return self.port

    async def stop(self) -> None:
        for pool in self._conns.values():
            for ss in pool:
                ss.close()
        self._conns.clear()
        for t in self._tasks:
            t.cancel()
        for t in self._tasks:
            try:
                await t
            except (asyncio.CancelledError, Exception):
                pass
        self._tasks.clear()
        if self._runner:
            await self._runner.cleanup()
            self._runner = None

    @web.middleware
    async def _log_middleware(self, request: web.Request, handler):
        # request logging with status + duration (reference gateway.go:107-135)
        t0 = time.monotonic()
        status = 500
        try:
            resp = await handler(request)
            status = resp.status
            return resp
        finally:
            self.log.info("%s %s %d %.1fms", request.method, request.path,
                          status, (time.monotonic() - t0) * 1e3)

    # ------------------------------------------------------------ discovery

    async def _discovery_loop(self) -> None:
        # gateway-side provider pull (reference gateway.go:351-423); unlike
        # the reference this shares the peer's single PeerManager code path
        # (deduplicate gateway-vs-peermanager discovery).
        while True:
            try:
                for res in await self.peer.discovery.discover_peers():
                    await self.peer.peer_manager.add_or_update_peer(res)
            except Exception as e:  # noqa: BLE001
                self.log.debug("gateway discovery failed: %s", e)
            await asyncio.sleep(self.cfg.intervals.gateway_discovery)

    # ------------------------------------------------------------- handlers

    def find_best_worker(self, model: str,
                         exclude: set[str] | None = None):
        return self.peer.peer_manager.find_best_worker(
            model, exclude=exclude,
            extra_load={k: 0.25 * v for k, v in self._inflight.items() if v})

    MAX_WORKER_ATTEMPTS = 3

    async def infer_with_failover(self, model: str, prompt: str):
        """Route to the best worker; on failure mark it and retry the
        next-best (extension — the reference errors out after its single
        best worker, gateway.go:200-214). Returns the PB response."""
        pm = self.peer.peer_manager
        tried: set[str] = set()
        last: Exception | None = None
        for _ in range(self.MAX_WORKER_ATTEMPTS):
            worker = self.find_best_worker(model, exclude=tried)
            if worker is None:
                break
            pid = worker.peer_id
            self._inflight[pid] = self._inflight.get(pid, 0) + 1
            try:
                return await self.request_inference(worker, model, prompt)
            except Exception as e:  # noqa: BLE001
                last = e
                tried.add(pid)
                pm.record_request_failure(pid)
                self.log.warning("worker %s failed (%s), trying next",
                                 pid, e)
            finally:
                self._inflight[pid] -= 1
        raise last or ConnectionError(f"no available worker for {model}")

    async def request_inference(self, worker, model: str, prompt: str,
                                stream: bool = False,
                                timeout: float = 300.0) -> pb.GenerateResponse:
        """Dial the worker's inference protocol, one PB request/response
        (reference RequestInference, gateway.go:243-293)."""
        addrs = list(worker.addrs)
        if not addrs:
            found = await self.peer.discovery.find_peer_addrs(worker.peer_id)
            addrs = found or []
        # pooled connection first (handshake amortization)
        pool = self._conns.setdefault(worker.peer_id, [])
        while pool:
            ss = pool.pop()
            if ss.is_closing():
                continue
            try:
                await ss.write_frame(
                    pb.request_message(model, prompt, stream).encode())
                frame = await ss.read_frame(timeout=timeout)
                resp = pb.BaseMessage.decode(frame).generate_response
                if resp is None:
                    raise ValueError("no GenerateResponse in reply")
                pool.append(ss)
                return resp
            except Exception:  # noqa: BLE001 — stale pooled conn; redial
                ss.close()
        last: Exception | None = None
        for addr in addrs:
            host, port = parse_addr(addr)
            try:
                # authenticated dial: fails unless the responder PROVES it
                # owns worker.peer_id (wire.py secure_open)
                ss = await secure_open(host, port, PROTO_INFERENCE,
                                       self.peer.identity,
                                       expected_peer_id=worker.peer_id)
                try:
                    await ss.write_frame(
                        pb.request_message(model, prompt, stream).encode())
                    frame = await ss.read_frame(timeout=timeout)
                except BaseException:
                    ss.close()
                    raise
                resp = pb.BaseMessage.decode(frame).generate_response
                if resp is None:
                    ss.close()
                    raise ValueError("no GenerateResponse in reply")
                pool.append(ss)      # keep for the next request
                return resp
            except Exception as e:  # noqa: BLE001
                last = e
        raise last or ConnectionError("worker unreachable")

    async def request_inference_stream(self, worker, model: str, prompt: str,
                                       timeout: float = 300.0):
        """Async iterator of GenerateResponse chunks from a streaming worker
        (extension; the reference is strictly one-shot, gateway.go:243-293)."""
        addrs = list(worker.addrs)
        if not addrs:
            addrs = await self.peer.discovery.find_peer_addrs(
                worker.peer_id) or []
        last: Exception | None = None
        for addr in addrs:
            host, port = parse_addr(addr)
            try:
                ss = await secure_open(host, port, PROTO_INFERENCE,
                                       self.peer.identity,
                                       expected_peer_id=worker.peer_id)
            except Exception as e:  # noqa: BLE001
                last = e
                continue
            try:
                await ss.write_frame(
                    pb.request_message(model, prompt, stream=True).encode())
                while True:
                    frame = await ss.read_frame(timeout=timeout)
                    resp = pb.BaseMessage.decode(frame).generate_response
                    if resp is None:
                        raise ValueError("no GenerateResponse in reply")
                    yield resp
                    if resp.done:
                        return
            finally:
                ss.close()
        raise last or ConnectionError("worker unreachable")

    async def _stream_ndjson(self, request: web.Request, worker, model: str,
                             prompt: str, chat: bool) -> web.StreamResponse:
        """Ollama-style streaming: one NDJSON object per chunk."""
        resp = web.StreamResponse(
            headers={"Content-Type": "application/x-ndjson"})
        await resp.prepare(request)
        pid = worker.peer_id
        self._inflight[pid] = self._inflight.get(pid, 0) + 1
        try:
            async for chunk in self.request_inference_stream(worker, model,
                                                             prompt):
                obj = {
                    "model": chunk.model,
                    "created_at": time.strftime("%Y-%m-%dT%H:%M:%SZ",
                                                time.gmtime()),
                    "done": chunk.done,
                }
                if chat:
                    obj["message"] = {"role": "assistant",
                                      "content": chunk.response}
                else:
                    obj["response"] = chunk.response
                if chunk.done:
                    obj["done_reason"] = chunk.done_reason or "stop"
                    obj["total_duration"] = chunk.total_duration
                    obj["worker_id"] = chunk.worker_id
                await resp.write(json.dumps(obj).encode("utf-8") + b"\n")
        except Exception as e:  # noqa: BLE001
            await resp.write(json.dumps(
                {"error": f"inference failed: {e}", "done": True}
            ).encode("utf-8") + b"\n")
        finally:
            self._inflight[pid] -= 1
        await resp.write_eof()
        return resp

    async def handle_chat(self, request: web.Request) -> web.Response:
        try:
            body = await request.json()
        except json.JSONDecodeError:
            return web.json_response({"error": "invalid JSON"}, status=400)
        model = body.get("model", "")
        messages = body.get("messages", [])
        if not model or not messages:
            return web.json_response(
                {"error": "model and messages are required"}, status=400)
        prompt = "\n".join(m.get("content", "") for m in messages
                           if isinstance(m, dict))
        worker = self.find_best_worker(model)
        if worker is None:
            return web.json_response(
                {"error": f"no available worker for model {model}"},
                status=503)  # gateway.go:192-199
        if body.get("stream"):
            return await self._stream_ndjson(request, worker, model, prompt,
                                             chat=True)
        try:
            resp = await self.infer_with_failover(model, prompt)
        except Exception as e:  # noqa: BLE001
            return web.json_response(
                {"error": f"inference failed: {e}"}, status=500)
        if resp.done_reason == "error":
            return web.json_response({"error": resp.response}, status=500)
        return web.json_response({
            "model": resp.model,
            "created_at": time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime()),
            "message": {"role": "assistant", "content": resp.response},
            "done": resp.done,
            "done_reason": resp.done_reason or "stop",
            "total_duration": resp.total_duration,
            "eval_count": resp.eval_count,
            "worker_id": resp.worker_id,
        })

    async def handle_generate(self, request: web.Request) -> web.Response:
        """Ollama /api/generate (prompt-in, response-out)."""
        try:
            body = await request.json()
        except json.JSONDecodeError:
            return web.json_response({"error": "invalid JSON"}, status=400)
        model = body.get("model", "")
        prompt = body.get("prompt", "")
        if not model:
            return web.json_response({"error": "model is required"},
                                     status=400)
        worker = self.find_best_worker(model)
        if worker is None:
            return web.json_response(
                {"error": f"no available worker for model {model}"},
                status=503)
        if body.get("stream"):
            return await self._stream_ndjson(request, worker, model, prompt,
                                             chat=False)
        try:
            resp = await self.infer_with_failover(model, prompt)
        except Exception as e:  # noqa: BLE001
            return web.json_response({"error": f"inference failed: {e}"},
                                     status=500)
        return web.json_response({
            "model": resp.model,
            "created_at": time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime()),
            "response": resp.response,
            "done": resp.done,
            "done_reason": resp.done_reason or "stop",
            "total_duration": resp.total_duration,
        })

    async def handle_health(self, request: web.Request) -> web.Response:
        """Full per-worker health map (reference gateway.go:426-461)."""
        pm = self.peer.peer_manager
        workers = []
        for pid, pi in pm.peers.items():
            r = pi.resource
            workers.append({
                "peer_id": pid,
                "healthy": pi.is_healthy,
                "worker_mode": r.worker_mode,
                "gpu_model": r.gpu_model,
                "supported_models": r.supported_models,
                "tokens_throughput": r.tokens_throughput,
                "vram_gb": r.vram_gb,
                "load": r.load,
                "last_seen": pi.last_seen,
                "failed_attempts": pi.failed_attempts,
            })
        return web.json_response({
            "status": "ok",
            "peer_id": self.peer.peer_id,
            "statistics": pm.get_peer_statistics(),
            "workers": workers,
        })

    async def handle_tags(self, request: web.Request) -> web.Response:
        """Ollama /api/tags analog: models available across the mesh."""
        models = set()
        for r in self.peer.peer_manager.get_healthy_peers():
            models.update(r.supported_models)
        return web.json_response(
            {"models": [{"name": m, "model": m} for m in sorted(models)]})
