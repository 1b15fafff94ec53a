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
