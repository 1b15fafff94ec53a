"""Configuration (reference parity: pkg/config/config.go:14-123).

Three layers: defaults <- env (CROWDLLAMA_*) <- CLI flags. Unlike the
reference, test-mode interval shrinking is a config field (`test_mode`)
rather than env-sniffing scattered through library code.
"""

from __future__ import annotations

import os
from dataclasses import dataclass, field


@dataclass
class Intervals:
    """All mesh timers in one place (reference constants cited inline)."""
    discovery: float = 10.0        # manager.go:69 DiscoveryInterval
    advertise: float = 1.0         # peer.go:455 1 s Provide loop
    metadata_update: float = 30.0  # peer.go:29 metadataUpdateInterval
    # main.go:266 5 s PublishMetadata ticker — subsumed here: metadata is
    # served inline over the metadata protocol (no CID publish step), so
    # this timer only exists for config parity.
    metadata_publish: float = 5.0
    health_check: float = 20.0     # manager.go:87 HealthCheckInterval
    stale_timeout: float = 60.0    # manager.go:85 PeerStaleTimeout
    cleanup: float = 20.0          # manager.go:522 cleanup loop
    max_failed_attempts: int = 3   # manager.go:89
    backoff_base: float = 10.0     # manager.go:90 linear backoff base
    metadata_timeout: float = 5.0  # manager.go:91 / peer.go 5 s deadline
    tombstone: float = 600.0       # manager.go:264 10 min tombstones
    metadata_max_age: float = 3600.0  # discovery.go:318 drop if stale > 1 h
    gateway_discovery: float = 10.0   # gateway.go:27
    stats_log: float = 10.0        # main.go:393
    nat_log: float = 30.0          # dht.go:279

    @classmethod
    def test_mode(cls) -> "Intervals":
        """Shrunk intervals (reference CROWDLLAMA_TEST_MODE=1 semantics)."""
        return cls(discovery=1.0, advertise=0.3, metadata_update=2.0,
                   metadata_publish=1.0, health_check=2.0, stale_timeout=20.0,
                   cleanup=2.0, backoff_base=1.0, metadata_timeout=3.0,
                   tombstone=30.0, gateway_discovery=0.5, stats_log=5.0,
                   nat_log=5.0)


@dataclass
class Config:
    verbose: bool = False
    key_path: str | None = None
    listen_host: str = "0.0.0.0"
    listen_port: int = 0            # 0 = ephemeral (reference: tcp/0)
    gateway_port: int = 9001        # gateway.go:25
    dht_port: int = 9000            # dht.go:25-28
    bootstrap_peers: list[str] = field(default_factory=list)  # "host:port"
    worker_mode: bool = False
    models: list[str] = field(default_factory=list)
    model_paths: dict[str, str] = field(default_factory=dict)
    engine: str = "hip"             # "hip" | "mock"
    test_mode: bool = False
    ipc_socket: str | None = None
    max_seq: int = 4096
    # every peer runs an embedded rendezvous server (reference: libp2p
    # DHT ModeServer on every peer)
    peer_dht: bool = True
    intervals: Intervals = field(default_factory=Intervals)

    def __post_init__(self):
        if self.test_mode:
            self.intervals = Intervals.test_mode()

    @classmethod
    def from_env(cls, **overrides) -> "Config":
        cfg = cls(**overrides)
        env = os.environ
        if env.get("CROWDLLAMA_VERBOSE", "") in ("1", "true", "yes"):
            cfg.verbose = True
        if env.get("CROWDLLAMA_KEY_PATH"):
            cfg.key_path = env["CROWDLLAMA_KEY_PATH"]
        if env.get("CROWDLLAMA_BOOTSTRAP"):
            cfg.bootstrap_peers = env["CROWDLLAMA_BOOTSTRAP"].split(",")
        if env.get("CROWDLLAMA_SOCKET"):
            cfg.ipc_socket = env["CROWDLLAMA_SOCKET"]
        if env.get("CROWDLLAMA_TEST_MODE") == "1":
            cfg.test_mode = True
            cfg.intervals = Intervals.test_mode()
        return cfg
