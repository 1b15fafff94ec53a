"""Peer identity keys (reference parity: internal/keys/keys.go:16-140).

Stable keyfile => stable PeerID across restarts. Keys are 32-byte random
seeds stored 0600 under ~/.crowdllama-amd/<component>.key; the peer id is a
base32 SHA-256 digest of the derived public value. (The reference uses
libp2p Ed25519 identities; signatures are not part of any reference data
path, so a digest identity preserves the observable semantics.)
"""

from __future__ import annotations

import base64
import hashlib
import os
import threading

_KEY_LOCK = threading.Lock()

COMPONENTS = ("dht", "worker", "consumer")


def default_key_path(component: str) -> str:
    base = os.environ.get("CROWDLLAMA_KEY_DIR",
                          os.path.join(os.path.expanduser("~"),
                                       ".crowdllama-amd"))
    return os.path.join(base, f"{component}.key")


def get_or_create_key(path: str) -> bytes:
    """Load (or create, 0600) a 32-byte identity seed."""
    with _KEY_LOCK:
        if os.path.exists(path):
            with open(path, "rb") as f:
                seed = f.read()
            if len(seed) != 32:
                raise ValueError(f"corrupt key file {path}: {len(seed)} bytes")
            return seed
        os.makedirs(os.path.dirname(path) or ".", mode=0o700, exist_ok=True)
        seed = os.urandom(32)
        fd = os.open(path, os.O_WRONLY | os.O_CREAT | os.O_EXCL, 0o600)
        with os.fdopen(fd, "wb") as f:
            f.write(seed)
        return seed


def peer_id_from_key(seed: bytes) -> str:
    pub = hashlib.sha256(b"crowdllama-amd-pub" + seed).digest()
    pid = hashlib.sha256(pub).digest()[:20]
    return "CLA" + base64.b32encode(pid).decode("ascii").rstrip("=")


def load_peer_id(component: str, key_path: str | None = None) -> tuple[str, bytes]:
    path = key_path or default_key_path(component)
    seed = get_or_create_key(path)
    return peer_id_from_key(seed), seed
