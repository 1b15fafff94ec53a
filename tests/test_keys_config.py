"""Key management + config tests (reference parity: keys_test.go,
config_test.go)."""

import os
import stat

import pytest

from crowdllama_amd.config import Config, Intervals
from crowdllama_amd.keys import get_or_create_key, load_peer_id, peer_id_from_key


def test_key_create_and_reload(tmp_path):
    path = str(tmp_path / "k" / "worker.key")
    k1 = get_or_create_key(path)
    assert len(k1) == 32
    assert os.path.exists(path)
    mode = stat.S_IMODE(os.stat(path).st_mode)
    assert mode == 0o600
    k2 = get_or_create_key(path)
    assert k1 == k2  # idempotent reload, byte-equal


def test_stable_peer_id(tmp_path):
    path = str(tmp_path / "worker.key")
    pid1, _ = load_peer_id("worker", path)
    pid2, _ = load_peer_id("worker", path)
    assert pid1 == pid2
    assert pid1.startswith("cla")  # hash of the ed25519 public key


def test_distinct_keys_distinct_ids():
    a = peer_id_from_key(b"a" * 32)
    b = peer_id_from_key(b"b" * 32)
    assert a != b


def test_corrupt_key_rejected(tmp_path):
    path = str(tmp_path / "bad.key")
    with open(path, "wb") as f:
        f.write(b"short")
    with pytest.raises(ValueError):
        get_or_create_key(path)


def test_config_defaults():
    cfg = Config()
    assert cfg.gateway_port == 9001   # reference gateway.go:25
    assert cfg.dht_port == 9000       # reference dht.go:25-28
    assert cfg.intervals.discovery == 10.0
    assert cfg.intervals.health_check == 20.0
    assert cfg.intervals.max_failed_attempts == 3
    assert cfg.intervals.stale_timeout == 60.0
    assert cfg.intervals.tombstone == 600.0


def test_config_test_mode_shrinks_intervals():
    cfg = Config(test_mode=True)
    assert cfg.intervals.discovery < Intervals().discovery
    assert cfg.intervals.health_check < Intervals().health_check


def test_config_from_env(monkeypatch):
    monkeypatch.setenv("CROWDLLAMA_VERBOSE", "1")
    monkeypatch.setenv("CROWDLLAMA_BOOTSTRAP", "10.0.0.1:9000,10.0.0.2:9000")
    monkeypatch.setenv("CROWDLLAMA_TEST_MODE", "1")
    cfg = Config.from_env()
    assert cfg.verbose is True
    assert cfg.bootstrap_peers == ["10.0.0.1:9000", "10.0.0.2:9000"]
    assert cfg.test_mode is True


def test_test_mode_shrinks_every_timer():
    """Guard: every interval field must actually shrink (or stay equal for
    counts) in test mode — catches fields added later but forgotten in
    Intervals.test_mode() (reference CROWDLLAMA_TEST_MODE semantics)."""
    from dataclasses import fields
    from crowdllama_amd.config import Intervals
    prod, test = Intervals(), Intervals.test_mode()
    for f in fields(Intervals):
        p, t = getattr(prod, f.name), getattr(test, f.name)
        if f.name == "max_failed_attempts":
            assert t == p  # a count, not a timer
        else:
            assert t <= p, f"{f.name}: test {t} > prod {p}"
