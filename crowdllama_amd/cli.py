"""CLI entry points (reference parity: cmd/crowdllama + cmd/dht — but a
first-party CLI rather than the reference's re-skinned Ollama cobra tree,
SURVEY.md §7.4).

    python -m crowdllama_amd.cli start [--worker-mode] [--models ...]
    python -m crowdllama_amd.cli dht [--port 9000]
    python -m crowdllama_amd.cli version
    python -m crowdllama_amd.cli network-status
"""

from __future__ import annotations

import argparse
import asyncio
import signal
import sys

from .config import Config
from .keys import load_peer_id
from .logutil import new_app_logger
from .version import version_string


def _add_common(ap: argparse.ArgumentParser) -> None:
    ap.add_argument("--verbose", action="store_true")
    ap.add_argument("--key", dest="key_path", default=None)
    ap.add_argument("--bootstrap", default="",
                    help="comma-separated host:port bootstrap nodes")
    ap.add_argument("--test-mode", action="store_true")


def _mk_config(args, **kw) -> Config:
    cfg = Config.from_env(**kw)
    cfg.verbose = cfg.verbose or args.verbose
    if args.key_path:
        cfg.key_path = args.key_path
    if args.bootstrap:
        cfg.bootstrap_peers = args.bootstrap.split(",")
    if not cfg.bootstrap_peers:
        cfg.bootstrap_peers = ["127.0.0.1:9000"]  # reference default
    if args.test_mode:
        cfg.test_mode = True
        from .config import Intervals
        cfg.intervals = Intervals.test_mode()
    return cfg


async def _wait_for_shutdown(log) -> None:
    loop = asyncio.get_running_loop()
    stop = asyncio.Event()
    for sig in (signal.SIGINT, signal.SIGTERM):
        try:
            loop.add_signal_handler(sig, stop.set)
        except NotImplementedError:
            pass
    await stop.wait()
    log.info("shutdown signal received")


async def _run_start(args) -> None:
    cfg = _mk_config(args)
    cfg.worker_mode = args.worker_mode
    cfg.gateway_port = args.port
    log = new_app_logger("crowdllama", cfg.verbose)
    log.info("%s", version_string())

    from .engine.api import MockEngine
    from .mesh.gateway import Gateway
    from .mesh.ipc import IPCServer
    from .mesh.peer import Peer

    engines = {}
    if args.worker_mode:
        models = [m for m in (args.models.split(",") if args.models else [])
                  if m]
        if args.engine == "mock":
            engines = {m: MockEngine(m) for m in (models or ["tinyllama"])}
        else:
            from .models import synth_path
            for i, m in enumerate(models or ["llama3-8b"]):
                path = (args.model_path or synth_path(m, scheme=args.scheme))
                log.info("loading %s from %s on device %d (batch=%d)", m,
                         path, args.device, args.batch)
                if args.batch > 1:
                    from .engine.batching import BatchingHipEngine
                    engines[m] = BatchingHipEngine(
                        m, path, device=args.device, batch=args.batch,
                        max_seq=cfg.max_seq)
                else:
                    from .engine.hip_engine import HipEngine
                    engines[m] = HipEngine(m, path, device=args.device,
                                           max_seq=cfg.max_seq)

    peer = Peer(cfg, worker_mode=args.worker_mode, engines=engines)
    await peer.start()

    gw = None
    if not args.worker_mode:
        gw = Gateway(peer, cfg)
        await gw.start()

    ipc = None
    if cfg.ipc_socket:
        async def ipc_handler(msg):
            from .mesh import pb
            req = msg.generate_request
            if req is None:
                return pb.response_message("", "Error: no request",
                                           done_reason="error")
            eng = engines.get(req.model)
            if eng is None:
                return pb.response_message(req.model,
                                           f"Error: model {req.model} not loaded",
                                           done_reason="error")
            result = await eng.generate(req.prompt)
            return pb.response_message(req.model, result.text, peer.peer_id)
        ipc = IPCServer(cfg, cfg.ipc_socket, ipc_handler)
        await ipc.start()

    # periodic stats logging (reference main.go:390-448)
    async def stats_loop():
        while True:
            await asyncio.sleep(cfg.intervals.stats_log)
            st = peer.peer_manager.get_peer_statistics()
            log.info("peers=%d healthy=%d workers=%d served=%d",
                     st["total_peers"], st["healthy_peers"], st["workers"],
                     peer.requests_served)
    stats = asyncio.create_task(stats_loop())
    try:
        await _wait_for_shutdown(log)
    finally:
        stats.cancel()
        if ipc:
            await ipc.stop()
        if gw:
            await gw.stop()
        await peer.stop()


async def _run_dht(args) -> None:
    cfg = _mk_config(args)
    cfg.dht_port = args.port
    log = new_app_logger("dht", cfg.verbose)
    log.info("%s", version_string())
    from .mesh.dhtnode import DHTServer
    peer_id, _ = load_peer_id("dht", cfg.key_path)
    srv = DHTServer(cfg, peer_id)
    port = await srv.start()
    log.info("bootstrap address: %s:%d", cfg.listen_host, port)
    try:
        await _wait_for_shutdown(log)
    finally:
        await srv.stop()


async def _run_network_status(args) -> None:
    cfg = _mk_config(args)
    from .keys import load_identity
    from .mesh.discovery import Discovery
    disco = Discovery(cfg.bootstrap_peers,
                      load_identity("consumer", cfg.key_path))
    ok = await disco.bootstrap_ok()
    print(f"bootstrap reachable: {ok}")
    if ok:
        for c in disco.clients:
            try:
                st = await c.stats()
                if st:
                    print(f"dht {c.addr}: peers={st['known_peers']} "
                          f"providers={st['providers']} "
                          f"conns={st['active_conns']}/{st['total_conns']} "
                          f"nat={st.get('nat', {})}")
                obs = await c.observed_addr()
                if obs:
                    print(f"observed address (via {c.addr}): {obs}")
            except Exception:  # noqa: BLE001
                pass
        peers = await disco.discover_peers()
        print(f"discovered {len(peers)} peers:")
        for r in peers:
            role = "worker" if r.worker_mode else "consumer"
            print(f"  {r.peer_id} [{role}] models={r.supported_models} "
                  f"{r.tokens_throughput:.0f} tok/s load={r.load:.2f} "
                  f"gpu={r.gpu_model}")
    await disco.close()


def main(argv: list[str] | None = None) -> int:
    ap = argparse.ArgumentParser(prog="crowdllama-amd")
    sub = ap.add_subparsers(dest="cmd", required=True)

    st = sub.add_parser("start", help="start a worker or consumer peer")
    _add_common(st)
    st.add_argument("--worker-mode", action="store_true")
    st.add_argument("--port", type=int, default=9001,
                    help="gateway HTTP port (consumer mode)")
    st.add_argument("--models", default="",
                    help="comma-separated model names to serve")
    st.add_argument("--model-path", default=None,
                    help="explicit GGUF path (single model)")
    st.add_argument("--scheme", default="q4_k_m")
    st.add_argument("--engine", default="hip", choices=["hip", "mock"])
    st.add_argument("--device", type=int, default=0)
    st.add_argument("--batch", type=int, default=1,
                    help="decode slots for continuous batching (worker mode)")

    dh = sub.add_parser("dht", help="start a bootstrap/rendezvous node")
    _add_common(dh)
    dh.add_argument("--port", type=int, default=9000)

    sub.add_parser("version")

    kg = sub.add_parser("keygen", help="create an identity key "
                        "(reference utils/dhtcertgen parity)")
    kg.add_argument("--component", default="dht",
                    choices=["dht", "worker", "consumer"])
    kg.add_argument("--out", default=None)

    ns = sub.add_parser("network-status")
    _add_common(ns)

    args = ap.parse_args(argv)
    if args.cmd == "version":
        print(version_string())
        return 0
    if args.cmd == "start":
        asyncio.run(_run_start(args))
        return 0
    if args.cmd == "dht":
        asyncio.run(_run_dht(args))
        return 0
    if args.cmd == "network-status":
        asyncio.run(_run_network_status(args))
        return 0
    if args.cmd == "keygen":
        from .keys import default_key_path, get_or_create_key, peer_id_from_key
        path = args.out or default_key_path(args.component)
        seed = get_or_create_key(path)
        print(f"{path}: peer id {peer_id_from_key(seed)}")
        return 0
    return 1


if __name__ == "__main__":
    sys.exit(main())
