"""IPC server for desktop UI bridge (reference parity: pkg/ipc/ipc.go).

Unix-domain socket at $CROWDLLAMA_SOCKET, chmod 0600. Each message is
sniffed by its first 4 bytes (reference ipc.go:187-240): a plausible
big-endian length (< 10 MB) selects the length-prefixed protobuf
BaseMessage path; anything else (JSON starts with '{' = 0x7B, which reads
as an implausibly large length) selects the newline-delimited JSON path
with message types ping/pong, initialize/initialize_status and
prompt/response (ipc.go:26-35). Both paths route prompts through the same
injected UnifiedAPIHandler (ipc.go:278-313,437-477)."""

from __future__ import annotations

import asyncio
import json
import os
import struct

from ..config import Config
from ..logutil import new_app_logger
from . import pb
from .wire import MAX_FRAME, write_frame

_MAX_JSON = 4096  # reference caps the JSON fallback read at 4 KB


class IPCServer:
    def __init__(self, cfg: Config, socket_path: str, handler):
        """handler: async (pb.BaseMessage) -> pb.BaseMessage."""
        self.cfg = cfg
        self.path = socket_path
        self.handler = handler
        self.log = new_app_logger("ipc", cfg.verbose)
        self._server: asyncio.base_events.Server | None = None

    async def start(self) -> None:
        if os.path.exists(self.path):
            os.unlink(self.path)
        os.makedirs(os.path.dirname(self.path) or ".", exist_ok=True)
        self._server = await asyncio.start_unix_server(self._on_conn,
                                                       self.path)
        os.chmod(self.path, 0o600)  # reference ipc.go:158
        self.log.info("IPC server on %s", self.path)

    async def stop(self) -> None:
        if self._server:
            self._server.close()
            try:   # bounded: 3.10 wait_closed blocks on live handlers
                await asyncio.wait_for(self._server.wait_closed(), 5)
            except asyncio.TimeoutError:
                pass
            self._server = None
        if os.path.exists(self.path):
            os.unlink(self.path)

    # ------------------------------------------------------------- PB path

    async def _handle_pb(self, frame: bytes, writer) -> None:
        msg = pb.BaseMessage.decode(frame)
        try:
            resp = await self.handler(msg)
        except Exception as e:  # noqa: BLE001
            resp = pb.response_message("", f"Error: {e}",
                                       done_reason="error")
        await write_frame(writer, resp.encode())

    # ----------------------------------------------------------- JSON path

    async def _handle_json(self, line: bytes, writer) -> None:
        try:
            msg = json.loads(line)
            mtype = msg.get("type", "")
        except (json.JSONDecodeError, AttributeError):
            await self._send_json(writer, {"type": "error",
                                           "error": "invalid JSON"})
            return
        if mtype == "ping":
            await self._send_json(writer, {"type": "pong"})
        elif mtype == "initialize":
            await self._send_json(writer, {"type": "initialize_status",
                                           "status": "ready"})
        elif mtype == "prompt":
            req = pb.request_message(msg.get("model", ""),
                                     msg.get("content", msg.get("prompt", "")))
            try:
                resp = await self.handler(req)
                gr = resp.generate_response
                await self._send_json(writer, {
                    "type": "response",
                    "model": gr.model if gr else "",
                    "content": gr.response if gr else "",
                    "done": gr.done if gr else True,
                })
            except Exception as e:  # noqa: BLE001
                await self._send_json(writer, {"type": "error",
                                               "error": str(e)})
        else:
            await self._send_json(writer, {"type": "error",
                                           "error": f"unknown type {mtype!r}"})

    @staticmethod
    async def _send_json(writer, obj: dict) -> None:
        writer.write(json.dumps(obj).encode("utf-8") + b"\n")
        await writer.drain()

    # ----------------------------------------------------------- conn loop

    async def _on_conn(self, reader: asyncio.StreamReader,
                       writer: asyncio.StreamWriter) -> None:
        try:
            while True:
                try:
                    head = await reader.readexactly(4)
                except (asyncio.IncompleteReadError, ConnectionError):
                    return
                (length,) = struct.unpack(">I", head)
                if length <= MAX_FRAME:
                    body = await reader.readexactly(length)
                    await self._handle_pb(body, writer)
                else:
                    rest = await reader.readline()
                    line = head + rest
                    if len(line) > _MAX_JSON:
                        await self._send_json(writer, {"type": "error",
                                                       "error": "too large"})
                        return
                    await self._handle_json(line, writer)
        except Exception as e:  # noqa: BLE001
            self.log.debug("ipc conn error: %s", e)
        finally:
            writer.close()
