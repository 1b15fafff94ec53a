"""Validate the tensor-parallel math (shard plan + all-reduce placement) by
simulating the engine's TP decode dataflow in numpy against the unsharded
reference. This is the CPU ground truth for the C++ TP path (which needs
multiple GPUs to run live)."""

import numpy as np

from crowdllama_amd.models.presets import ModelConfig
from crowdllama_amd.parallel.tp import shard_plan


def _rms(x, g, eps=1e-5):
    return x / np.sqrt((x * x).mean() + eps) * g


def _softmax(x):
    e = np.exp(x - x.max())
    return e / e.sum()


def test_tp_layer_dataflow_matches_full():
    rng = np.random.default_rng(0)
    cfg = ModelConfig("tp-test", vocab_size=512, hidden_size=512, n_layers=1,
                      n_heads=8, n_kv_heads=2, ffn_hidden=1024,
                      rope_theta=10000.0)
    tp = 2
    h, f, hd = cfg.hidden_size, cfg.ffn_hidden, cfg.head_dim
    nh, nkv = cfg.n_heads, cfg.n_kv_heads
    G = nh // nkv
    W = {
        "attn_norm.weight": rng.standard_normal(h).astype(np.float32),
        "ffn_norm.weight": rng.standard_normal(h).astype(np.float32),
        "attn_q.weight": rng.standard_normal((h, h)).astype(np.float32) * 0.05,
        "attn_k.weight": rng.standard_normal((nkv * hd, h)).astype(np.float32) * 0.05,
        "attn_v.weight": rng.standard_normal((nkv * hd, h)).astype(np.float32) * 0.05,
        "attn_output.weight": rng.standard_normal((h, h)).astype(np.float32) * 0.05,
        "ffn_gate.weight": rng.standard_normal((f, h)).astype(np.float32) * 0.05,
        "ffn_up.weight": rng.standard_normal((f, h)).astype(np.float32) * 0.05,
        "ffn_down.weight": rng.standard_normal((h, f)).astype(np.float32) * 0.05,
    }
    x = rng.standard_normal(h).astype(np.float32)

    def attn(q, k, v, nheads, kvheads):
        # single token, len-1 kv: softmax over one position = v
        out = np.zeros((nheads, hd), dtype=np.float32)
        for head in range(nheads):
            kvh = head // (nheads // kvheads)
            s = np.array([k[kvh] @ q[head] / np.sqrt(hd)])
            out[head] = _softmax(s) @ v[kvh][None, :]
        return out

    # ---- full (unsharded) ----
    xn = _rms(x, W["attn_norm.weight"])
    q = (W["attn_q.weight"] @ xn).reshape(nh, hd)
    k = (W["attn_k.weight"] @ xn).reshape(nkv, hd)
    v = (W["attn_v.weight"] @ xn).reshape(nkv, hd)
    ao = attn(q, k, v, nh, nkv).reshape(-1)
    x1 = x + W["attn_output.weight"] @ ao
    xn2 = _rms(x1, W["ffn_norm.weight"])
    g = W["ffn_gate.weight"] @ xn2
    u = W["ffn_up.weight"] @ xn2
    act = (g / (1 + np.exp(-g))) * u
    x2_full = x1 + W["ffn_down.weight"] @ act

    # ---- sharded (engine dataflow: local partials + all-reduce) ----
    attn_partials, ffn_partials = [], []
    for rank in range(tp):
        plan = shard_plan(cfg, rank, tp)
        def sl(name):
            s_ = plan[name]
            return W[name][s_.r0:s_.r1, s_.c0:s_.c1]
        xn_l = _rms(x, W["attn_norm.weight"])  # replicated norm
        q_l = (sl("attn_q.weight") @ xn_l).reshape(nh // tp, hd)
        k_l = (sl("attn_k.weight") @ xn_l).reshape(nkv // tp, hd)
        v_l = (sl("attn_v.weight") @ xn_l).reshape(nkv // tp, hd)
        ao_l = attn(q_l, k_l, v_l, nh // tp, nkv // tp).reshape(-1)
        part = sl("attn_output.weight") @ ao_l
        if rank == 0:
            part = part + x  # rank 0 folds the residual
        attn_partials.append(part)
    x1_tp = np.sum(attn_partials, axis=0)  # all-reduce
    np.testing.assert_allclose(x1_tp, x1, rtol=2e-5, atol=2e-5)

    for rank in range(tp):
        plan = shard_plan(cfg, rank, tp)
        def sl(name):
            s_ = plan[name]
            return W[name][s_.r0:s_.r1, s_.c0:s_.c1]
        xn2_l = _rms(x1_tp, W["ffn_norm.weight"])
        g_l = sl("ffn_gate.weight") @ xn2_l
        u_l = sl("ffn_up.weight") @ xn2_l
        act_l = (g_l / (1 + np.exp(-g_l))) * u_l
        part = sl("ffn_down.weight") @ act_l
        if rank == 0:
            part = part + x1_tp
        ffn_partials.append(part)
    x2_tp = np.sum(ffn_partials, axis=0)
    np.testing.assert_allclose(x2_tp, x2_full, rtol=2e-5, atol=2e-5)


def test_tp_head_allgather_layout():
    """Vocab-sharded head + slice-offset all-gather reproduces full logits."""
    rng = np.random.default_rng(1)
    V, h, tp = 64, 32, 4
    Whead = rng.standard_normal((V, h)).astype(np.float32)
    xn = rng.standard_normal(h).astype(np.float32)
    full = Whead @ xn
    logits = np.zeros(V, dtype=np.float32)
    per = V // tp
    for rank in range(tp):
        logits[rank * per:(rank + 1) * per] = Whead[rank * per:(rank + 1) * per] @ xn
    np.testing.assert_allclose(logits, full, rtol=1e-6)
