"""Mesh security: ed25519/X25519 primitives (RFC test vectors), the
authenticated handshake, encrypted framing, and record signatures.

Reference bar: the reference gets noise/TLS encryption
and cryptographically-verified peer IDs from libp2p
(internal/keys/keys.go:38-98, internal/discovery/discovery.go:48-84);
these tests pin the MI355X mesh's equivalents.
"""

import asyncio
import os

import pytest

from crowdllama_amd.keys import identity_from_seed
from crowdllama_amd.mesh import crypto
from crowdllama_amd.mesh.resource import Resource


def test_ed25519_rfc8032_vectors():
    seed = bytes.fromhex(
        "9d61b19deffd5a60ba844af492ec2cc44449c5697b326919703bac031cae7f60")
    pub = crypto.ed25519_public(seed)
    assert pub.hex() == ("d75a980182b10ab7d54bfed3c964073a"
                         "0ee172f3daa62325af021a68f707511a")
    sig = crypto.ed25519_sign(seed, b"")
    assert sig.hex() == (
        "e5564300c360ac729086e2cc806e828a84877f1eb8e5d974d873e06522490155"
        "5fb8821590a33bacc61e39701cf9b46bd25bf5f0595bbe24655141438e7a100b")
    assert crypto.ed25519_verify(pub, b"", sig)
    assert not crypto.ed25519_verify(pub, b"tampered", sig)
    bad = bytearray(sig)
    bad[0] ^= 1
    assert not crypto.ed25519_verify(pub, b"", bytes(bad))


def test_x25519_rfc7748_vector():
    k = bytes.fromhex(
        "a546e36bf0527c9d3b16154b82465edd62144c0ac1fc5a18506a2244ba449ac4")
    u = bytes.fromhex(
        "e6db6867583030db3594c1a424b15f7c726624ec26b3353b10a903a6d0ab1c4c")
    assert crypto.x25519(k, u).hex() == (
        "c3da55379de9c6908e94ea4df28d084f32eccf03491c71f754b4075577a28552")


def test_dh_agreement_and_channel():
    ea, eb = os.urandom(32), os.urandom(32)
    assert crypto.x25519(ea, crypto.x25519_public(eb)) == \
        crypto.x25519(eb, crypto.x25519_public(ea))
    ia = identity_from_seed(b"a" * 32)
    ib = identity_from_seed(b"b" * 32)
    m1 = crypto.handshake_msg1(ea, ia.pub)
    m2 = crypto.handshake_msg2(eb, ib.seed, ib.pub, m1)
    chA = crypto.derive_channels(ea, m2[:32], m1, m2, initiator=True)
    chB = crypto.derive_channels(eb, m1[:32], m1, m2, initiator=False)
    msg = os.urandom(10000)
    assert chB.open(chA.seal(msg)) == msg
    assert chA.open(chB.seal(b"reply")) == b"reply"
    # replay rejection: re-opening an already-consumed frame fails (the
    # receive counter advanced)
    f2 = chA.seal(b"second")
    assert chB.open(f2) == b"second"
    with pytest.raises(ValueError):
        chB.open(f2)
    # tamper detection (fresh pair: a failed frame desynchronizes the
    # strictly-ordered channel by design)
    chA2 = crypto.derive_channels(ea, m2[:32], m1, m2, initiator=True)
    chB2 = crypto.derive_channels(eb, m1[:32], m1, m2, initiator=False)
    f = bytearray(chA2.seal(b"x"))
    f[0] ^= 1
    with pytest.raises(ValueError):
        chB2.open(bytes(f))


def test_secure_stream_roundtrip_and_identity():
    from crowdllama_amd.mesh import wire

    async def go():
        server_id = identity_from_seed(b"s" * 32)
        client_id = identity_from_seed(b"c" * 32)
        got = {}

        async def on_conn(reader, writer):
            try:
                ss, proto = await wire.secure_accept(reader, writer,
                                                     server_id)
                got["proto"] = proto
                got["client"] = ss.peer_id
                data = await ss.read_frame(timeout=5)
                await ss.write_frame(data[::-1])
            finally:
                writer.close()

        srv = await asyncio.start_server(on_conn, "127.0.0.1", 0)
        port = srv.sockets[0].getsockname()[1]
        try:
            ss = await wire.secure_open(
                "127.0.0.1", port, "/t/1.0.0", client_id,
                expected_peer_id=server_id.peer_id)
            assert ss.peer_id == server_id.peer_id
            await ss.write_frame(b"abc")
            assert await ss.read_frame(timeout=5) == b"cba"
            ss.close()
            assert got["proto"] == "/t/1.0.0"
            assert got["client"] == client_id.peer_id

            # dialing with the WRONG expected identity must fail closed
            with pytest.raises(wire.WireError):
                await wire.secure_open(
                    "127.0.0.1", port, "/t/1.0.0", client_id,
                    expected_peer_id=identity_from_seed(b"z" * 32).peer_id)
        finally:
            srv.close()
            await srv.wait_closed()
    asyncio.run(go())


def test_resource_signature():
    ident = identity_from_seed(b"w" * 32)
    r = Resource(peer_id=ident.peer_id, supported_models=["m"],
                 tokens_throughput=42.0, worker_mode=True,
                 addrs=["127.0.0.1:1"])
    r.touch()
    r.sign(ident.seed, ident.pub)
    assert r.verify()
    rt = Resource.from_json(r.to_json())
    assert rt.verify()

    # tampered field -> invalid
    rt.tokens_throughput = 9999.0
    assert not rt.verify()

    # forged peer_id: signed by a key that does NOT hash to the id
    other = identity_from_seed(b"x" * 32)
    forged = Resource(peer_id=ident.peer_id, supported_models=["m"],
                      worker_mode=True)
    forged.touch()
    forged.sign(other.seed, other.pub)
    assert not forged.verify()

    # unsigned record -> invalid
    bare = Resource(peer_id=ident.peer_id)
    bare.touch()
    assert not bare.verify()
