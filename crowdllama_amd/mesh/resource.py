"""Peer metadata record (reference parity: pkg/crowdllama/types.go:30-74).

Unlike the reference — which hardcodes "RTX 4090"/150 tok/s/24 GB
(peer.go:319-343) — workers here fill these from real HIP device props and a
measured rolling throughput (SURVEY.md §7.3 "throughput metadata honesty").
"""

from __future__ import annotations

import json
import time
from dataclasses import dataclass, field

from . import crypto


@dataclass
class Resource:
    peer_id: str = ""
    supported_models: list[str] = field(default_factory=list)
    tokens_throughput: float = 0.0   # tokens/sec (measured, rolling)
    vram_gb: float = 0.0
    load: float = 0.0                # 0..1
    gpu_model: str = ""
    last_updated: float = 0.0        # unix seconds
    version: str = ""
    worker_mode: bool = False
    addrs: list[str] = field(default_factory=list)  # "host:port"
    # this peer's embedded rendezvous server (reference parity: every
    # libp2p peer runs the DHT in ModeServer, pkg/dht/dht.go:106-112);
    # consumers add it as a rendezvous target, so losing the bootstrap
    # node leaves N live servers, not one gossip fallback
    dht_addr: str = ""
    # record authentication: pubkey is the advertising
    # peer's ed25519 public key (hex); sig signs the canonical payload.
    # Receivers verify sig AND that peer_id == hash(pubkey), so a record
    # cannot be forged for another peer's id. (The reference relies on
    # libp2p stream authentication instead; records here also travel via
    # rendezvous, so they carry their own proof.)
    pubkey: str = ""
    sig: str = ""

    def touch(self) -> None:
        self.last_updated = time.time()

    def _payload(self) -> dict:
        return {
            "peer_id": self.peer_id,
            "supported_models": self.supported_models,
            "tokens_throughput": self.tokens_throughput,
            "vram_gb": self.vram_gb,
            "load": self.load,
            "gpu_model": self.gpu_model,
            "last_updated": self.last_updated,
            "version": self.version,
            "worker_mode": self.worker_mode,
            "addrs": self.addrs,
            "dht_addr": self.dht_addr,
        }

    def _signing_bytes(self) -> bytes:
        return json.dumps(self._payload(), sort_keys=True,
                          separators=(",", ":")).encode("utf-8")

    def sign(self, seed: bytes, pub: bytes) -> None:
        self.pubkey = pub.hex()
        self.sig = crypto.ed25519_sign(seed, self._signing_bytes()).hex()

    def verify(self) -> bool:
        """True iff the record is signed by the key that owns peer_id."""
        try:
            pub = bytes.fromhex(self.pubkey)
            sig = bytes.fromhex(self.sig)
        except ValueError:
            return False
        if crypto.peer_id_from_pub(pub) != self.peer_id:
            return False
        return crypto.ed25519_verify(pub, self._signing_bytes(), sig)

    def to_json(self) -> str:
        d = self._payload()
        d["pubkey"] = self.pubkey
        d["sig"] = self.sig
        return json.dumps(d)

    @classmethod
    def from_json(cls, data: str | bytes) -> "Resource":
        d = json.loads(data)
        if not isinstance(d, dict) or "peer_id" not in d:
            raise ValueError("invalid resource JSON")
        r = cls()
        r.peer_id = d.get("peer_id", "")
        r.supported_models = list(d.get("supported_models", []))
        r.tokens_throughput = float(d.get("tokens_throughput", 0.0))
        r.vram_gb = float(d.get("vram_gb", 0.0))
        r.load = float(d.get("load", 0.0))
        r.gpu_model = d.get("gpu_model", "")
        r.last_updated = float(d.get("last_updated", 0.0))
        r.version = d.get("version", "")
        r.worker_mode = bool(d.get("worker_mode", False))
        r.addrs = list(d.get("addrs", []))
        r.dht_addr = d.get("dht_addr", "")
        r.pubkey = d.get("pubkey", "")
        r.sig = d.get("sig", "")
        return r

    def age(self) -> float:
        return time.time() - self.last_updated
