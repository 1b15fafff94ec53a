"""IPC server for desktop UI bridge (reference parity: pkg/ipc/ipc.go).

Unix-domain socket at $CROWDLLAMA_SOCKET, chmod 0600, speaking
length-prefixed protobuf BaseMessage only (the reference's JSON/PB sniffing
dual protocol is deliberately dropped — SURVEY.md §7.4; message semantics
kept: prompt -> UnifiedAPIHandler -> response)."""

from __future__ import annotations

import asyncio
import os

from ..config import Config
from ..logutil import new_app_logger
from . import pb
from .wire import read_frame, write_frame


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
            await self._server.wait_closed()
            self._server = None
        if os.path.exists(self.path):
            os.unlink(self.path)

    async def _on_conn(self, reader: asyncio.StreamReader,
                       writer: asyncio.StreamWriter) -> None:
        try:
            while True:
                try:
                    frame = await read_frame(reader)
                except (asyncio.IncompleteReadError, ConnectionError):
                    return
                msg = pb.BaseMessage.decode(frame)
                try:
                    resp = await self.handler(msg)
                except Exception as e:  # noqa: BLE001
                    resp = pb.response_message("", f"Error: {e}",
                                               done_reason="error")
                await write_frame(writer, resp.encode())
        except Exception as e:  # noqa: BLE001
            self.log.debug("ipc conn error: %s", e)
        finally:
            writer.close()
