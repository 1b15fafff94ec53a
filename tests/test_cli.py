"""CLI end-to-end on loopback: the actual `python -m crowdllama_amd.cli`
entry points wired together as subprocesses (reference parity: the
cmd/crowdllama + cmd/dht binaries of SURVEY.md §2.1, exercised the way the
reference's integration test drives its built binary)."""

import json
import socket
import subprocess
import sys
import time
import urllib.request

import pytest


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _spawn(args, tmp_path, name):
    return subprocess.Popen(
        [sys.executable, "-m", "crowdllama_amd.cli", *args],
        stdout=(tmp_path / f"{name}.out").open("wb"),
        stderr=subprocess.STDOUT)


def test_cli_version():
    out = subprocess.run([sys.executable, "-m", "crowdllama_amd.cli",
                          "version"], capture_output=True, text=True)
    assert out.returncode == 0
    assert "crowdllama-amd" in out.stdout


def test_cli_keygen(tmp_path):
    key = tmp_path / "id.key"
    out = subprocess.run([sys.executable, "-m", "crowdllama_amd.cli",
                          "keygen", "--out", str(key)],
                         capture_output=True, text=True)
    assert out.returncode == 0
    assert key.exists()
    assert "peer id cla" in out.stdout


def _start_mesh(tmp_path, attempt):
    """Spawn dht+worker+consumer on freshly probed ports. Returns
    (procs, gw_port, boot) or Nones if a process lost the port race."""
    dht_port = _free_port()
    gw_port = _free_port()
    boot = f"127.0.0.1:{dht_port}"
    procs = [_spawn(["dht", "--port", str(dht_port), "--test-mode",
                     "--key", str(tmp_path / f"dht{attempt}.key")],
                    tmp_path, "dht")]
    time.sleep(0.5)
    procs.append(_spawn(["start", "--worker-mode", "--engine", "mock",
                         "--models", "m1", "--test-mode",
                         "--bootstrap", boot,
                         "--key", str(tmp_path / f"w{attempt}.key")],
                        tmp_path, "worker"))
    procs.append(_spawn(["start", "--test-mode", "--bootstrap", boot,
                         "--port", str(gw_port),
                         "--key", str(tmp_path / f"c{attempt}.key")],
                        tmp_path, "consumer"))
    time.sleep(0.5)
    if any(p.poll() is not None for p in procs):
        # a process lost the probe->bind port race; caller retries
        for p in procs:
            p.terminate()
        for p in procs:
            try:
                p.wait(timeout=10)
            except subprocess.TimeoutExpired:
                p.kill()
        return None, None, None
    return procs, gw_port, boot


@pytest.mark.timeout(180)
def test_cli_mesh_end_to_end(tmp_path):
    procs = None
    for attempt in range(3):
        procs, gw_port, boot = _start_mesh(tmp_path, attempt)
        if procs is not None:
            break
    assert procs is not None, "could not bind mesh ports in 3 attempts"
    try:
        # poll health until the worker is discovered
        deadline = time.time() + 60
        found = False
        while time.time() < deadline and not found:
            for p in procs:
                assert p.poll() is None, \
                    (tmp_path / "dht.out").read_text() + \
                    (tmp_path / "worker.out").read_text() + \
                    (tmp_path / "consumer.out").read_text()
            try:
                with urllib.request.urlopen(
                        f"http://127.0.0.1:{gw_port}/api/health",
                        timeout=2) as r:
                    health = json.load(r)
                found = any(w["healthy"] and "m1" in w["supported_models"]
                            for w in health.get("workers", []))
            except Exception:
                pass
            if not found:
                time.sleep(0.3)
        assert found, "worker never appeared in gateway health"
        req = urllib.request.Request(
            f"http://127.0.0.1:{gw_port}/api/chat",
            data=json.dumps({"model": "m1", "messages": [
                {"role": "user", "content": "hi"}]}).encode(),
            headers={"Content-Type": "application/json"})
        with urllib.request.urlopen(req, timeout=30) as r:
            body = json.load(r)
        assert body["done"] is True
        assert "mock response" in body["message"]["content"]
        # network-status against the live mesh
        out = subprocess.run([sys.executable, "-m", "crowdllama_amd.cli",
                              "network-status", "--bootstrap", boot,
                              "--test-mode"],
                             capture_output=True, text=True, timeout=30)
        assert out.returncode == 0
        assert "bootstrap reachable: True" in out.stdout
        assert "worker" in out.stdout
    finally:
        for p in procs:
            p.terminate()
        for p in procs:
            try:
                p.wait(timeout=10)
            except subprocess.TimeoutExpired:
                p.kill()


@pytest.mark.gpu
@pytest.mark.timeout(180)
def test_cli_mesh_hip_engine(tmp_path):
    """Full CLI path with the real HIP engine (+continuous batching) on one
    GPU: `cla start --worker-mode --engine hip --batch 4`."""
    from crowdllama_amd.models import synth_path
    path = synth_path("testllama", scheme="q4_k_m", mode="exact", seed=7)
    procs = None
    for attempt in range(3):
        dht_port = _free_port()
        gw_port = _free_port()
        boot = f"127.0.0.1:{dht_port}"
        procs = [_spawn(["dht", "--port", str(dht_port), "--test-mode",
                         "--key", str(tmp_path / f"dht{attempt}.key")],
                        tmp_path, "dht")]
        time.sleep(0.5)
        procs.append(_spawn(["start", "--worker-mode", "--engine", "hip",
                             "--models", "testllama", "--model-path", path,
                             "--batch", "4", "--test-mode",
                             "--bootstrap", boot,
                             "--key", str(tmp_path / f"w{attempt}.key")],
                            tmp_path, "worker"))
        procs.append(_spawn(["start", "--test-mode", "--bootstrap", boot,
                             "--port", str(gw_port),
                             "--key", str(tmp_path / f"c{attempt}.key")],
                            tmp_path, "consumer"))
        time.sleep(0.5)
        if any(p.poll() is not None for p in procs):  # port race: retry
            for p in procs:
                p.terminate()
            for p in procs:
                try:
                    p.wait(timeout=10)
                except subprocess.TimeoutExpired:
                    p.kill()
            procs = None
            continue
        break
    assert procs is not None, "could not bind mesh ports in 3 attempts"
    try:
        deadline = time.time() + 120
        found = False
        while time.time() < deadline and not found:
            for p in procs:
                assert p.poll() is None, \
                    (tmp_path / "worker.out").read_text()[-2000:]
            try:
                with urllib.request.urlopen(
                        f"http://127.0.0.1:{gw_port}/api/health",
                        timeout=2) as r:
                    health = json.load(r)
                found = any(w["healthy"] and
                            "testllama" in w["supported_models"] and
                            ("gfx" in w["gpu_model"] or
                             "MI" in w["gpu_model"].upper())
                            for w in health.get("workers", []))
            except Exception:
                pass
            if not found:
                time.sleep(0.3)
        assert found, "HIP worker never appeared in gateway health"
        req = urllib.request.Request(
            f"http://127.0.0.1:{gw_port}/api/chat",
            data=json.dumps({"model": "testllama", "messages": [
                {"role": "user", "content": "abc"}]}).encode(),
            headers={"Content-Type": "application/json"})
        with urllib.request.urlopen(req, timeout=60) as r:
            body = json.load(r)
        assert body["done"] is True
        assert isinstance(body["message"]["content"], str)
    finally:
        for p in procs:
            p.terminate()
        for p in procs:
            try:
                p.wait(timeout=10)
            except subprocess.TimeoutExpired:
                p.kill()
