"""Test helpers (reference parity: pkg/testhelpers/testhelpers.go —
CreateIsolatedTestDHT + deterministic test ports)."""

from __future__ import annotations

import contextlib
import hashlib

from .config import Config
from .mesh.dhtnode import DHTServer


def get_test_port(test_name: str) -> int:
    """Deterministic port in 10000-14999 from the test name (reference
    testhelpers.go:63-71 uses an FNV hash; any stable hash works)."""
    h = int.from_bytes(hashlib.sha1(test_name.encode()).digest()[:4], "big")
    return 10000 + h % 5000


@contextlib.asynccontextmanager
async def isolated_test_dht(port: int = 0, peer_id: str = "CLATESTDHT"):
    """Loopback DHT server for tests (reference testhelpers.go:19-60)."""
    cfg = Config(test_mode=True, listen_host="127.0.0.1")
    srv = DHTServer(cfg, peer_id)
    bound = await srv.start("127.0.0.1", port)
    try:
        yield srv, f"127.0.0.1:{bound}"
    finally:
        await srv.stop()
