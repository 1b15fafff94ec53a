"""IPC server tests (reference parity: ipc_test.go — real unix socket,
mock handler injected at the API-handler seam)."""

import asyncio

from crowdllama_amd.config import Config
from crowdllama_amd.mesh import pb
from crowdllama_amd.mesh.ipc import IPCServer
from crowdllama_amd.mesh.wire import read_frame, write_frame


def test_ipc_prompt_roundtrip(tmp_path):
    async def go():
        sock = str(tmp_path / "cla.sock")
        calls = []

        async def handler(msg):
            calls.append(msg)
            req = msg.generate_request
            return pb.response_message(req.model, f"echo:{req.prompt}", "w0")

        cfg = Config(test_mode=True)
        srv = IPCServer(cfg, sock, handler)
        await srv.start()
        try:
            reader, writer = await asyncio.open_unix_connection(sock)
            req = pb.request_message("m1", "hello ipc")
            await write_frame(writer, req.encode())
            frame = await read_frame(reader, timeout=5.0)
            resp = pb.BaseMessage.decode(frame).generate_response
            assert resp is not None
            assert resp.response == "echo:hello ipc"
            assert resp.worker_id == "w0"
            # second message on the same connection
            await write_frame(writer, pb.request_message("m1", "two").encode())
            frame = await read_frame(reader, timeout=5.0)
            assert pb.BaseMessage.decode(frame).generate_response.response == \
                "echo:two"
            writer.close()
            assert len(calls) == 2
        finally:
            await srv.stop()
    asyncio.run(go())


def test_ipc_handler_error_stringified(tmp_path):
    async def go():
        sock = str(tmp_path / "cla.sock")

        async def handler(msg):
            raise ValueError("boom")

        srv = IPCServer(Config(test_mode=True), sock, handler)
        await srv.start()
        try:
            reader, writer = await asyncio.open_unix_connection(sock)
            await write_frame(writer, pb.request_message("m", "x").encode())
            frame = await read_frame(reader, timeout=5.0)
            resp = pb.BaseMessage.decode(frame).generate_response
            assert resp.done_reason == "error"
            assert "boom" in resp.response
            writer.close()
        finally:
            await srv.stop()
    asyncio.run(go())


def test_ipc_json_fallback(tmp_path):
    """Reference dual protocol (ipc.go:187-240): the same socket accepts
    newline-delimited JSON — ping/pong, initialize, prompt/response."""
    async def go():
        sock = str(tmp_path / "cla.sock")

        async def handler(msg):
            req = msg.generate_request
            return pb.response_message(req.model, f"echo:{req.prompt}", "w0")

        cfg = Config(test_mode=True)
        srv = IPCServer(cfg, sock, handler)
        await srv.start()
        try:
            import json
            reader, writer = await asyncio.open_unix_connection(sock)

            async def rpc(obj):
                writer.write(json.dumps(obj).encode() + b"\n")
                await writer.drain()
                return json.loads(await asyncio.wait_for(reader.readline(), 5))

            assert (await rpc({"type": "ping"}))["type"] == "pong"
            init = await rpc({"type": "initialize"})
            assert init["type"] == "initialize_status"
            resp = await rpc({"type": "prompt", "model": "m1",
                              "content": "hi"})
            assert resp["type"] == "response"
            assert resp["content"] == "echo:hi"
            bad = await rpc({"type": "nope"})
            assert bad["type"] == "error"
            # PB on the same socket, fresh connection
            r2, w2 = await asyncio.open_unix_connection(sock)
            await write_frame(w2, pb.request_message("m1", "pb msg").encode())
            frame = await read_frame(r2, timeout=5.0)
            assert pb.BaseMessage.decode(frame).generate_response.response == \
                "echo:pb msg"
            w2.close()
            writer.close()
        finally:
            await srv.stop()
    asyncio.run(go())
