"""Length-prefixed framing + protocol negotiation over asyncio streams.

Framing parity with the reference (pkg/crowdllama/pbwire.go:14-70): 4-byte
big-endian length prefix, 10 MB cap. Every mesh connection opens with a
length-prefixed UTF-8 protocol id (the analog of libp2p protocol selection),
then speaks that protocol's frames.
"""

from __future__ import annotations

import asyncio
import json
import struct
from typing import Any

MAX_FRAME = 10 * 1024 * 1024  # reference cap: pbwire.go:53

# protocol ids (reference: pkg/crowdllama/types.go:17-27)
PROTO_INFERENCE = "/crowdllama-amd/inference/1.0.0"
PROTO_METADATA = "/crowdllama-amd/metadata/1.0.0"
PROTO_RENDEZVOUS = "/crowdllama-amd/rendezvous/1.0.0"
NAMESPACE = "crowdllama-ns"  # reference: types.go:27


class WireError(Exception):
    pass


async def write_frame(writer: asyncio.StreamWriter, payload: bytes) -> None:
    if len(payload) > MAX_FRAME:
        raise WireError(f"frame too large: {len(payload)}")
    writer.write(struct.pack(">I", len(payload)) + payload)
    await writer.drain()


async def read_frame(reader: asyncio.StreamReader,
                     timeout: float | None = None) -> bytes:
    async def _read():
        hdr = await reader.readexactly(4)
        (n,) = struct.unpack(">I", hdr)
        if n > MAX_FRAME:
            raise WireError(f"frame too large: {n}")
        return await reader.readexactly(n)
    if timeout is None:
        return await _read()
    return await asyncio.wait_for(_read(), timeout)


async def write_json(writer: asyncio.StreamWriter, obj: Any) -> None:
    await write_frame(writer, json.dumps(obj).encode("utf-8"))


async def read_json(reader: asyncio.StreamReader,
                    timeout: float | None = None) -> Any:
    return json.loads((await read_frame(reader, timeout)).decode("utf-8"))


async def open_protocol(host: str, port: int, proto: str,
                        timeout: float = 5.0):
    """Dial a peer and negotiate a protocol. Returns (reader, writer)."""
    reader, writer = await asyncio.wait_for(
        asyncio.open_connection(host, port), timeout)
    await write_frame(writer, proto.encode("utf-8"))
    return reader, writer


async def accept_protocol(reader: asyncio.StreamReader,
                          timeout: float = 5.0) -> str:
    """Server side: read the requested protocol id."""
    return (await read_frame(reader, timeout)).decode("utf-8")
