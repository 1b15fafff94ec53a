"""TP shard-plan and sampling unit tests (CPU)."""

import numpy as np
import pytest

from crowdllama_amd.engine.sampling import sample
from crowdllama_amd.models import get_preset
from crowdllama_amd.parallel.tp import local_meta, shard_plan, validate_tp


@pytest.mark.parametrize("model,tp", [("llama3-70b", 8), ("llama3-8b", 8),
                                      ("llama3-8b", 4), ("mistral-7b", 2)])
def test_shard_plan_covers(model, tp):
    cfg = get_preset(model)
    validate_tp(cfg, tp)
    # row shards tile the full row space; col shards tile cols
    for name in ["attn_q.weight", "ffn_gate.weight", "output.weight"]:
        rows = set()
        for r in range(tp):
            s = shard_plan(cfg, r, tp)[name]
            assert s.kind == "rows"
            rows.update(range(s.r0, s.r1))
        full = {"attn_q.weight": cfg.hidden_size,
                "ffn_gate.weight": cfg.ffn_hidden,
                "output.weight": cfg.vocab_size}[name]
        assert rows == set(range(full))
    for name in ["attn_output.weight", "ffn_down.weight"]:
        cols = set()
        for r in range(tp):
            s = shard_plan(cfg, r, tp)[name]
            assert s.kind == "cols"
            assert s.c0 % 256 == 0  # quant superblock alignment
            cols.update(range(s.c0, s.c1))
        full = cfg.hidden_size if name == "attn_output.weight" else cfg.ffn_hidden
        assert cols == set(range(full))


def test_shard_alignment_rejects_bad_tp():
    cfg = get_preset("tinyllama")  # 4 kv heads
    with pytest.raises(ValueError):
        validate_tp(cfg, 8)


def test_local_meta():
    cfg = get_preset("llama3-70b")
    m = local_meta(cfg, 8)
    assert m["heads"] == 8 and m["kv_heads"] == 1
    assert m["vocab_shard"] == 128256 // 8


def test_sample_greedy():
    logits = np.array([0.1, 5.0, -1.0, 4.9])
    assert sample(logits, temperature=0.0) == 1


def test_sample_temperature_distribution():
    rng = np.random.default_rng(0)
    logits = np.array([2.0, 1.0, 0.0, -10.0])
    counts = np.zeros(4)
    for _ in range(500):
        counts[sample(logits, temperature=1.0, rng=rng)] += 1
    assert counts[0] > counts[1] > counts[2]
    assert counts[3] == 0 or counts[3] < 5


def test_sample_top_k():
    rng = np.random.default_rng(1)
    logits = np.array([3.0, 2.0, 1.0, 0.5])
    seen = {sample(logits, temperature=1.0, top_k=2, rng=rng)
            for _ in range(200)}
    assert seen <= {0, 1}


def test_sample_top_p():
    rng = np.random.default_rng(2)
    logits = np.array([10.0, 9.0, -5.0, -5.0])
    seen = {sample(logits, temperature=1.0, top_p=0.9, rng=rng)
            for _ in range(200)}
    assert seen <= {0, 1}


def test_rolling_rate_window():
    """RollingRate reports tokens/sec over its sliding window (replaces the
    reference's hardcoded 150 tok/s advertisement, peer.go:323)."""
    import time as _t
    from crowdllama_amd.engine.api import RollingRate
    rr = RollingRate(window=0.4)
    assert rr.rate() == 0.0
    rr.add(100)
    _t.sleep(0.05)
    rr.add(100)
    r = rr.rate()
    assert r > 100.0, r  # 200 tokens over well under a second
    _t.sleep(0.6)
    rr.add(0)  # trigger pruning
    assert rr.rate() < 10.0  # old events fell out of the window


def test_mock_engine_stream_matches_generate():
    """The default streaming seam reassembles to the one-shot result."""
    import asyncio
    from crowdllama_amd.engine.api import MockEngine

    async def go():
        eng = MockEngine("m", response="alpha beta gamma")
        full = await eng.generate("x")
        parts = []
        async for c in eng.generate_stream("x"):
            parts.append(c)
        assert "".join(p.text for p in parts) == full.text
        assert parts[-1].done_reason == "stop"
        assert all(not p.done_reason for p in parts[:-1])
    asyncio.run(go())
