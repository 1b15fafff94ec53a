"""Wire-format tests (reference parity: pbwire_test.go, types_test.go)."""

import asyncio

import pytest

from crowdllama_amd.mesh import pb
from crowdllama_amd.mesh.resource import Resource
from crowdllama_amd.mesh.wire import (MAX_FRAME, WireError, read_frame,
                                      write_frame)


def test_generate_request_roundtrip():
    msg = pb.request_message("llama3-8b", "hello world", stream=True)
    raw = msg.encode()
    back = pb.BaseMessage.decode(raw)
    assert back.generate_request is not None
    assert back.generate_response is None
    assert back.generate_request.model == "llama3-8b"
    assert back.generate_request.prompt == "hello world"
    assert back.generate_request.stream is True


def test_generate_response_roundtrip():
    msg = pb.response_message("m", "resp text", worker_id="w1",
                              done_reason="stop", total_duration_ns=12345)
    back = pb.BaseMessage.decode(msg.encode())
    r = back.generate_response
    assert r is not None
    assert r.model == "m"
    assert r.response == "resp text"
    assert r.done is True
    assert r.worker_id == "w1"
    assert r.total_duration == 12345
    assert r.created_at.seconds > 0


def test_empty_fields_roundtrip():
    m = pb.BaseMessage(generate_request=pb.GenerateRequest())
    back = pb.BaseMessage.decode(m.encode())
    assert back.generate_request is not None
    assert back.generate_request.model == ""
    assert back.generate_request.stream is False


def test_unicode_prompt():
    msg = pb.request_message("m", "héllo wörld 你好 🚀")
    back = pb.BaseMessage.decode(msg.encode())
    assert back.generate_request.prompt == "héllo wörld 你好 🚀"


def test_varint_edge_cases():
    from crowdllama_amd.mesh.pb import _dec_varint, _enc_varint
    for v in [0, 1, 127, 128, 300, 2**32, 2**63 - 1]:
        enc = _enc_varint(v)
        dec, pos = _dec_varint(enc, 0)
        assert dec == v and pos == len(enc)


def test_frame_roundtrip():
    async def run():
        reader = asyncio.StreamReader()
        transport_data = bytearray()

        class W:
            def write(self, b):
                transport_data.extend(b)
            async def drain(self):
                pass
        await write_frame(W(), b"hello")
        reader.feed_data(bytes(transport_data))
        got = await read_frame(reader)
        assert got == b"hello"
    asyncio.run(run())


def test_frame_too_large():
    async def run():
        class W:
            def write(self, b):
                pass
            async def drain(self):
                pass
        with pytest.raises(WireError):
            await write_frame(W(), b"x" * (MAX_FRAME + 1))
    asyncio.run(run())


def test_resource_json_roundtrip():
    r = Resource(peer_id="CLAXYZ", supported_models=["llama3-8b"],
                 tokens_throughput=1234.5, vram_gb=288.0, load=0.25,
                 gpu_model="AMD Instinct MI355X", worker_mode=True,
                 addrs=["127.0.0.1:1234"])
    r.touch()
    back = Resource.from_json(r.to_json())
    assert back.peer_id == "CLAXYZ"
    assert back.supported_models == ["llama3-8b"]
    assert back.tokens_throughput == 1234.5
    assert back.vram_gb == 288.0
    assert back.worker_mode is True
    assert back.addrs == ["127.0.0.1:1234"]
    assert back.age() < 5.0


def test_resource_invalid_json():
    with pytest.raises(Exception):
        Resource.from_json("not json")
    with pytest.raises(ValueError):
        Resource.from_json("[1,2,3]")


def test_frame_fragmented_delivery():
    """Frames reassemble across arbitrary TCP fragmentation."""
    async def run():
        reader = asyncio.StreamReader()
        buf = bytearray()

        class W:
            def write(self, b):
                buf.extend(b)
            async def drain(self):
                pass
        payload = bytes(range(256)) * 5
        await write_frame(W(), payload)
        # feed one byte at a time
        task = asyncio.ensure_future(read_frame(reader))
        for i in range(len(buf)):
            reader.feed_data(bytes(buf[i:i + 1]))
            await asyncio.sleep(0)
        got = await task
        assert got == payload
    asyncio.run(run())


def test_two_frames_back_to_back():
    async def run():
        reader = asyncio.StreamReader()
        buf = bytearray()

        class W:
            def write(self, b):
                buf.extend(b)
            async def drain(self):
                pass
        await write_frame(W(), b"first")
        await write_frame(W(), b"second")
        reader.feed_data(bytes(buf))
        assert await read_frame(reader) == b"first"
        assert await read_frame(reader) == b"second"
    asyncio.run(run())


def test_protocol_negotiation_loopback():
    """open_protocol/accept_protocol handshake over a real socket, and an
    unknown protocol is surfaced to the acceptor (reference: libp2p stream
    protocol IDs, types.go:12-27)."""
    from crowdllama_amd.mesh.wire import (PROTO_INFERENCE, accept_protocol,
                                          open_protocol)

    async def run():
        seen = []

        async def on_conn(reader, writer):
            seen.append(await accept_protocol(reader))
            writer.close()

        server = await asyncio.start_server(on_conn, "127.0.0.1", 0)
        port = server.sockets[0].getsockname()[1]
        try:
            _, w = await open_protocol("127.0.0.1", port, PROTO_INFERENCE)
            w.close()
            _, w = await open_protocol("127.0.0.1", port, "/bogus/9.9")
            w.close()
            await asyncio.sleep(0.2)
            assert seen[0] == PROTO_INFERENCE
            assert seen[1] == "/bogus/9.9"   # acceptor sees and can reject
        finally:
            server.close()
            await server.wait_closed()
    asyncio.run(run())


def test_unknown_fields_skipped():
    """proto3 forward compatibility: decoders skip unknown field numbers so
    newer peers can extend the schema without breaking old ones."""
    from crowdllama_amd.mesh.pb import _enc_varint

    base = pb.request_message("m1", "hello").encode()
    # splice unknown fields into the inner GenerateRequest payload:
    # re-encode by appending to the OUTER message instead (same skip path)
    extra = (_enc_varint(99 << 3 | 0) + _enc_varint(12345)           # varint
             + _enc_varint((100 << 3) | 2) + _enc_varint(3) + b"xyz")  # bytes
    back = pb.BaseMessage.decode(base + extra)
    assert back.generate_request is not None
    assert back.generate_request.model == "m1"
    assert back.generate_request.prompt == "hello"


def test_read_frame_timeout():
    """read_frame enforces its deadline on a silent peer (the worker's 5 s
    inference read deadline — reference peer.go:259-271)."""
    async def run():
        reader = asyncio.StreamReader()  # nothing ever fed
        with pytest.raises(asyncio.TimeoutError):
            await read_frame(reader, timeout=0.2)
    asyncio.run(run())


def test_resource_defaults_isolated():
    """Mutable defaults must not be shared between Resource instances
    (reference types_test.go checks field defaults)."""
    a = Resource(peer_id="A")
    b = Resource(peer_id="B")
    a.supported_models.append("m")
    a.addrs.append("x")
    assert b.supported_models == []
    assert b.addrs == []
    assert a.age() >= 0.0
