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


# --------------------------------------------------------------- security
# Authenticated, encrypted streams (crypto.py): X25519 ephemeral key
# agreement bound to ed25519 identities, then every frame sealed
# (encrypt-then-MAC, per-direction keys, strict counters). The reference
# gets the equivalent from libp2p's noise/TLS security transports
# (reference internal/discovery/discovery.go:48-84); round 1 spoke
# plaintext TCP and trusted any claimed peer_id.

import os as _os

from . import crypto


class SecureStream:
    """Framed, encrypted, mutually-authenticated stream."""

    def __init__(self, reader: asyncio.StreamReader,
                 writer: asyncio.StreamWriter, chan: "crypto.SecureChannel",
                 peer_id: str, peer_pub: bytes):
        self._r = reader
        self._w = writer
        self._ch = chan
        self.peer_id = peer_id        # cryptographically verified
        self.peer_pub = peer_pub

    async def write_frame(self, payload: bytes) -> None:
        await write_frame(self._w, self._ch.seal(payload))

    async def read_frame(self, timeout: float | None = None) -> bytes:
        return self._ch.open(await read_frame(self._r, timeout))

    async def write_json(self, obj: Any) -> None:
        await self.write_frame(json.dumps(obj).encode("utf-8"))

    async def read_json(self, timeout: float | None = None) -> Any:
        return json.loads((await self.read_frame(timeout)).decode("utf-8"))

    def close(self) -> None:
        try:
            self._w.close()
        except Exception:
            pass

    def is_closing(self) -> bool:
        return self._w.is_closing()


async def secure_open(host: str, port: int, proto: str, ident,
                      timeout: float = 5.0,
                      expected_peer_id: str | None = None) -> SecureStream:
    """Dial, run the handshake as initiator, negotiate `proto` inside the
    tunnel. With expected_peer_id set, the connection fails unless the
    responder PROVES that identity (its ed25519 key hashes to the id and
    it signed the ephemeral exchange)."""
    reader, writer = await asyncio.wait_for(
        asyncio.open_connection(host, port), timeout)
    try:
        eph = _os.urandom(32)
        m1 = crypto.handshake_msg1(eph, ident.pub)
        await write_frame(writer, m1)
        m2 = await read_frame(reader, timeout)
        if len(m2) != 128:
            raise WireError("bad handshake response")
        e_r, pub_r, sig_r = m2[:32], m2[32:64], m2[64:]
        if not crypto.ed25519_verify(
                pub_r, crypto.PROTO_TAG + b"|resp|" + m1 + e_r + pub_r,
                sig_r):
            raise WireError("responder signature invalid")
        rid = crypto.peer_id_from_pub(pub_r)
        if expected_peer_id is not None and rid != expected_peer_id:
            raise WireError(
                f"peer identity mismatch: dialed {expected_peer_id}, "
                f"got {rid}")
        await write_frame(writer, crypto.handshake_msg3(ident.seed, m1, m2))
        chan = crypto.derive_channels(eph, e_r, m1, m2, initiator=True)
        ss = SecureStream(reader, writer, chan, rid, pub_r)
        await ss.write_frame(proto.encode("utf-8"))
        return ss
    except BaseException:
        writer.close()
        raise


async def secure_accept(reader: asyncio.StreamReader,
                        writer: asyncio.StreamWriter, ident,
                        timeout: float = 5.0) -> tuple[SecureStream, str]:
    """Server side: run the handshake as responder (authenticating the
    initiator too), then read the negotiated protocol id."""
    m1 = await read_frame(reader, timeout)
    if len(m1) != 64:
        raise WireError("bad handshake init")
    e_i, pub_i = m1[:32], m1[32:64]
    eph = _os.urandom(32)
    m2 = crypto.handshake_msg2(eph, ident.seed, ident.pub, m1)
    await write_frame(writer, m2)
    m3 = await read_frame(reader, timeout)
    if not crypto.ed25519_verify(
            pub_i, crypto.PROTO_TAG + b"|init|" + m1 + m2, m3):
        raise WireError("initiator signature invalid")
    chan = crypto.derive_channels(eph, e_i, m1, m2, initiator=False)
    ss = SecureStream(reader, writer, chan, crypto.peer_id_from_pub(pub_i),
                      pub_i)
    proto = (await ss.read_frame(timeout)).decode("utf-8")
    return ss, proto
