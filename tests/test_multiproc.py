"""Multi-process distributed paths on CPU (gloo, world_size=2): the
bench.py rendezvous/barrier/all-reduce code the driver exercises with one
rank per GPU."""

import json
import os
import socket
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def test_bench_rendezvous_world2(tmp_path):
    """bench.py under torch.distributed.run with 2 CPU ranks: ranks must
    rendezvous, barrier, and exit cleanly (no GPU -> null result line)."""
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    # dynamic port (a fixed one collides with orphaned rendezvous stores
    # from interrupted runs); retry once — the probe->bind window can still
    # lose a race with other processes on a busy box
    lines = []
    r = None
    for _ in range(2):
        port = str(_free_port())
        r = subprocess.run(
            [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
             "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
             "--master-port", port, "bench.py", "--gpus", "2",
             "--steps", "2", "--warmup", "0", "--model", "testllama",
             "--scheme", "q8_0"],
            cwd=REPO, env=env, capture_output=True, text=True, timeout=300)
        lines = [ln for ln in r.stdout.splitlines() if ln.startswith("{")]
        if r.returncode == 0 and lines:
            break
    assert r.returncode == 0, r.stderr[-2000:]
    assert lines, f"no JSON output: {r.stdout[-500:]} {r.stderr[-500:]}"
    d = json.loads(lines[-1])
    assert d["value"] is None and "no GPU" in d.get("error", "")


def test_gloo_allreduce_max():
    """The MAX-over-ranks reduction bench.py uses, in-process (world=1)."""
    import torch
    import torch.distributed as dist
    if dist.is_initialized():
        pytest.skip("process group already active")
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(_free_port())
    dist.init_process_group("gloo", rank=0, world_size=1)
    try:
        t = torch.tensor([1.5], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        assert t.item() == 1.5
    finally:
        dist.destroy_process_group()
