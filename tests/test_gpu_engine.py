"""End-to-end GPU engine vs the pure-numpy fp32 reference, on a tiny
random-init GGUF (exact-mode quantization so both sides read identical
weights)."""

import numpy as np
import pytest

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def core():
    from crowdllama_amd.ops import get_core
    c = get_core()
    if c.device_count() == 0:
        pytest.skip("no GPU")
    return c


@pytest.fixture(scope="module")
def tiny_gguf(tmp_path_factory):
    from crowdllama_amd.models import write_synthetic_gguf
    path = str(tmp_path_factory.mktemp("m") / "tiny.gguf")
    write_synthetic_gguf(path, "testllama", scheme="q4_k_m", mode="exact",
                         seed=7)
    return path


def test_engine_logits_vs_ref(core, tiny_gguf):
    from crowdllama_amd.engine.ref_numpy import RefLlama
    cfg = core.EngineConfig()
    cfg.batch = 1
    cfg.max_seq = 128
    cfg.act_q8 = False  # strict comparison against the f32 reference
    eng = core.Engine(tiny_gguf, cfg)
    prompt = [3, 17, 99, 250, 7]
    ids = np.array([prompt], dtype=np.int32)
    eng.prefill(ids)
    got = np.asarray(eng.logits(0))

    ref = RefLlama(tiny_gguf)
    logits = None
    for t in prompt:
        logits = ref.step(t)
    assert got.shape == logits.shape
    # bf16 KV + fp32 accumulation differences: compare tightly but not exactly
    denom = np.abs(logits).max() + 1e-6
    rel = np.abs(got - logits).max() / denom
    assert rel < 5e-3, f"max rel err {rel}"
    assert int(np.argmax(got)) == int(np.argmax(logits))


def test_engine_greedy_matches_ref(core, tiny_gguf):
    from crowdllama_amd.engine.ref_numpy import RefLlama
    cfg = core.EngineConfig()
    cfg.batch = 1
    cfg.max_seq = 128
    cfg.act_q8 = False
    eng = core.Engine(tiny_gguf, cfg)
    prompt = [5, 10, 200]
    n_new = 8
    eng.prefill(np.array([prompt], dtype=np.int32))
    eng.decode(n_new - 1)
    got = list(eng.gen_tokens(0))

    ref = RefLlama(tiny_gguf)
    want = ref.generate(prompt, n_new)
    # Greedy chains can diverge after an early near-tie; require a matching
    # prefix of at least 4 tokens and identical first token.
    assert got[0] == want[0]
    match = 0
    for a, b in zip(got, want):
        if a != b:
            break
        match += 1
    assert match >= 4, f"got {got} want {want}"


def test_engine_gemv_r_shapes(core, tmp_path_factory):
    """B=1 logits parity on a model whose projections hit the register-x
    GEMV (k_gemv_r: K=2048 half-stripe for qkv/o/gate_up/head, K=4096
    full stripe for down) instead of the tiny-shape legacy kernel the
    other tests cover — including the o-projection residual prefetch."""
    from crowdllama_amd.engine.ref_numpy import RefLlama
    from crowdllama_amd.models import write_synthetic_gguf
    from crowdllama_amd.models.presets import ModelConfig
    mc = ModelConfig("gemvr", vocab_size=512, hidden_size=2048, n_layers=2,
                     n_heads=16, n_kv_heads=8, ffn_hidden=4096,
                     rope_theta=10000.0, max_seq_len=512)
    path = str(tmp_path_factory.mktemp("m") / "gemvr.gguf")
    write_synthetic_gguf(path, mc, scheme="q4_k_m", mode="exact", seed=11)
    cfg = core.EngineConfig()
    cfg.batch = 1
    cfg.max_seq = 128
    cfg.act_q8 = False
    eng = core.Engine(path, cfg)
    prompt = [3, 17, 99, 250, 7]
    eng.prefill(np.array([prompt], dtype=np.int32))
    got = np.asarray(eng.logits(0))
    ref = RefLlama(path)
    logits = None
    for t in prompt:
        logits = ref.step(t)
    denom = np.abs(logits).max() + 1e-6
    rel = np.abs(got - logits).max() / denom
    # this hidden-2048 config sits at ~5e-3 vs the f32 reference (bf16
    # KV + split-K atomicAdd ordering); observed 0.0051 on some boxes,
    # so the bound leaves margin — argmax + greedy prefix carry the
    # semantic check
    assert rel < 1.5e-2, f"max rel err {rel}"
    assert int(np.argmax(got)) == int(np.argmax(logits))
    # decode steps run every projection through k_gemv_r (the prefill
    # above only exercised the head GEMV)
    eng.decode(5)
    got_ids = list(eng.gen_tokens(0))
    want = ref.generate(prompt, 6)
    # the logits comparison above is the primary check; greedy chains on
    # synthetic weights can flip at near-ties (split-K atomicAdd order
    # varies run to run), so require only the first token + a short
    # prefix here
    assert got_ids[0] == want[0]
    match = 0
    for a, b in zip(got_ids, want):
        if a != b:
            break
        match += 1
    assert match >= 2, f"got {got_ids} want {want}"


def test_engine_attn_single_split(core, tiny_gguf):
    """The fence-free S=1 attention combine (no fan-in, no ticket, no
    agent-scope L2 invalidate — the B>16 default) produces the same
    logits as the reference. CLA_ATTN_SPLITS is read at engine init."""
    import os
    from crowdllama_amd.engine.ref_numpy import RefLlama
    cfg = core.EngineConfig()
    cfg.batch = 1
    cfg.max_seq = 128
    cfg.act_q8 = False
    os.environ["CLA_ATTN_SPLITS"] = "1"
    try:
        eng = core.Engine(tiny_gguf, cfg)
    finally:
        del os.environ["CLA_ATTN_SPLITS"]
    prompt = [3, 17, 99, 250, 7]
    eng.prefill(np.array([prompt], dtype=np.int32))
    eng.decode(4)
    got = np.asarray(eng.logits(0))
    ref = RefLlama(tiny_gguf)
    logits = None
    for t in prompt:
        logits = ref.step(t)
    for t in list(eng.gen_tokens(0))[:4]:
        logits = ref.step(t)
    denom = np.abs(logits).max() + 1e-6
    rel = np.abs(got - logits).max() / denom
    # teacher-forced, but bf16-KV rounding accumulates over the 5 steps
    assert rel < 1e-2, f"max rel err {rel}"


def test_engine_batch2(core, tiny_gguf):
    cfg = core.EngineConfig()
    cfg.batch = 2
    cfg.max_seq = 128
    eng = core.Engine(tiny_gguf, cfg)  # act_q8 default: the serving path
    prompts = np.array([[3, 17, 99], [3, 17, 99]], dtype=np.int32)
    eng.prefill(prompts)
    eng.decode(5)
    a = list(eng.gen_tokens(0))
    b = list(eng.gen_tokens(1))
    assert a == b, f"identical prompts must generate identically: {a} vs {b}"


def test_engine_reset(core, tiny_gguf):
    cfg = core.EngineConfig()
    cfg.batch = 1
    eng = core.Engine(tiny_gguf, cfg)
    eng.prefill(np.array([[1, 2, 3]], dtype=np.int32))
    eng.decode(3)
    first = list(eng.gen_tokens(0))
    eng.reset()
    eng.prefill(np.array([[1, 2, 3]], dtype=np.int32))
    eng.decode(3)
    second = list(eng.gen_tokens(0))
    assert first == second


def test_engine_long_context_page_crossing(core, tiny_gguf):
    """KV spans multiple 64-token pages; logits still match the reference."""
    from crowdllama_amd.engine.ref_numpy import RefLlama
    import numpy as np
    cfg = core.EngineConfig()
    cfg.batch = 1
    cfg.max_seq = 256
    cfg.act_q8 = False
    eng = core.Engine(tiny_gguf, cfg)
    rng = np.random.default_rng(3)
    prompt = rng.integers(3, 500, size=150).tolist()  # crosses 2+ pages
    eng.prefill(np.asarray([prompt], dtype=np.int32))
    got = np.asarray(eng.logits(0))
    ref = RefLlama(tiny_gguf)
    want = None
    for t in prompt:
        want = ref.step(t)
    rel = np.abs(got - want).max() / (np.abs(want).max() + 1e-9)
    assert rel < 1e-2, rel
    assert int(np.argmax(got)) == int(np.argmax(want))


def test_engine_midrange_prompt_splitk(core, tiny_gguf):
    """Prompt length in 33..128: prefill GEMM chunks hit BM=32 tiles WITH
    split-K enabled — the range where round 1's gemm_uses_splitk /
    launch_gemm_ex divergence skipped the C pre-zero and accumulated into
    stale scratch (advisor finding; fixed by gemm_splitk_factor)."""
    from crowdllama_amd.engine.ref_numpy import RefLlama
    import numpy as np
    cfg = core.EngineConfig()
    cfg.batch = 1
    cfg.max_seq = 256
    cfg.act_q8 = False
    eng = core.Engine(tiny_gguf, cfg)
    rng = np.random.default_rng(11)
    prompt = rng.integers(3, 500, size=70).tolist()
    eng.prefill(np.asarray([prompt], dtype=np.int32))
    got = np.asarray(eng.logits(0))
    ref = RefLlama(tiny_gguf)
    want = None
    for t in prompt:
        want = ref.step(t)
    rel = np.abs(got - want).max() / (np.abs(want).max() + 1e-9)
    assert rel < 1e-2, rel
    assert int(np.argmax(got)) == int(np.argmax(want))

    # back-to-back prefills must not leak split-K partials between passes
    # (tolerance, not bitwise: atomicAdd accumulation order varies per run)
    eng.reset()
    eng.prefill(np.asarray([prompt], dtype=np.int32))
    again = np.asarray(eng.logits(0))
    rel2 = np.abs(again - got).max() / (np.abs(got).max() + 1e-9)
    assert rel2 < 5e-3, rel2  # stale-scratch corruption would be O(1)


def test_engine_slot_parking(core, tiny_gguf):
    """Parked slots do not advance during shared decode steps; unparked
    slots decode identically to a fresh engine (serving slot lifecycle)."""
    import numpy as np
    prompt = [3, 17, 99]
    cfg = core.EngineConfig()
    cfg.batch = 2
    cfg.max_seq = 64
    cfg.act_q8 = False  # compare against the B=1 f32 GEMV path below
    eng = core.Engine(tiny_gguf, cfg)
    eng.prefill(np.asarray([prompt, prompt], dtype=np.int32))
    eng.set_slot_active(1, False)
    eng.decode(5)
    active_toks = list(eng.gen_tokens(0))
    parked = eng.n_past()
    assert parked[1] == len(prompt), "parked slot advanced during decode"
    assert len(active_toks) == 6  # prefill token + 5 decode steps

    cfg1 = core.EngineConfig()
    cfg1.batch = 1
    cfg1.max_seq = 64
    cfg1.act_q8 = False
    e1 = core.Engine(tiny_gguf, cfg1)
    e1.prefill(np.asarray([prompt], dtype=np.int32))
    e1.decode(5)
    assert active_toks == list(e1.gen_tokens(0))


def test_engine_bf16_scheme(core, tmp_path_factory):
    """bf16 weights (the 70B TP dtype) through the same engine."""
    from crowdllama_amd.engine.ref_numpy import RefLlama
    from crowdllama_amd.models import write_synthetic_gguf
    import numpy as np
    path = str(tmp_path_factory.mktemp("bf") / "bf.gguf")
    write_synthetic_gguf(path, "testllama", scheme="bf16", mode="exact",
                         seed=13)
    cfg = core.EngineConfig()
    cfg.batch = 1
    cfg.max_seq = 64
    eng = core.Engine(path, cfg)
    prompt = [5, 9, 44]
    eng.prefill(np.asarray([prompt], dtype=np.int32))
    got = np.asarray(eng.logits(0))
    ref = RefLlama(path)
    want = None
    for t in prompt:
        want = ref.step(t)
    rel = np.abs(got - want).max() / (np.abs(want).max() + 1e-9)
    assert rel < 1e-2, rel


def test_engine_batched_gemm_path(core, tiny_gguf):
    """B=4 decode exercises the split-K MFMA path; identical prompts in
    every slot must generate identically (and match the GEMV path)."""
    import numpy as np
    prompt = [3, 17, 99]
    cfg1 = core.EngineConfig()
    cfg1.batch = 1
    cfg1.max_seq = 64
    cfg1.act_q8 = False  # f32-path equivalence (i8 has its own tests)
    e1 = core.Engine(tiny_gguf, cfg1)
    e1.prefill(np.asarray([prompt], dtype=np.int32))
    e1.decode(5)
    ref_tokens = list(e1.gen_tokens(0))

    cfg4 = core.EngineConfig()
    cfg4.batch = 4
    cfg4.max_seq = 64
    cfg4.act_q8 = False
    e4 = core.Engine(tiny_gguf, cfg4)
    e4.prefill(np.asarray([prompt] * 4, dtype=np.int32))
    e4.decode(5)
    for slot in range(4):
        assert list(e4.gen_tokens(slot)) == ref_tokens, slot


def test_engine_act_q8_decode(core, tiny_gguf):
    """Default serving path: int8-quantized activations for quantized
    weights (v_dot4). Logits stay close to the f32 reference and the
    engine's argmax lands in the reference's top-5."""
    from crowdllama_amd.engine.ref_numpy import RefLlama
    import numpy as np
    cfg = core.EngineConfig()
    cfg.batch = 1
    cfg.max_seq = 128
    cfg.act_q8 = True  # exercise the optional int8-activation path
    eng = core.Engine(tiny_gguf, cfg)
    prompt = [3, 17, 99, 250, 7]
    eng.prefill(np.asarray([prompt], dtype=np.int32))
    got = np.asarray(eng.logits(0))
    ref = RefLlama(tiny_gguf, act_q8=True)
    want = None
    for t in prompt:
        want = ref.step(t)
    rel = np.abs(got - want).max() / (np.abs(want).max() + 1e-9)
    assert rel < 2e-2, rel
    top5 = np.argsort(want)[-5:]
    assert int(np.argmax(got)) in top5


def test_engine_act_q8_batched_gemm(core, tiny_gguf):
    """Batched decode (B=4, GEMM path) with act_q8: runs the int8-MFMA
    GEMM (gemm_i8.hip) for quantized projections. Logits stay close to the
    act_q8 numpy reference; all slots with identical prompts agree."""
    from crowdllama_amd.engine.ref_numpy import RefLlama
    import numpy as np
    cfg = core.EngineConfig()
    cfg.batch = 4
    cfg.max_seq = 128
    cfg.act_q8 = True
    eng = core.Engine(tiny_gguf, cfg)
    prompt = [3, 17, 99, 250, 7]
    eng.prefill(np.asarray([prompt] * 4, dtype=np.int32))
    got = np.asarray(eng.logits(0))
    for slot in range(1, 4):
        other = np.asarray(eng.logits(slot))
        np.testing.assert_allclose(other, got, rtol=1e-4, atol=1e-4)
    ref = RefLlama(tiny_gguf, act_q8=True)
    want = None
    for t in prompt:
        want = ref.step(t)
    rel = np.abs(got - want).max() / (np.abs(want).max() + 1e-9)
    assert rel < 2e-2, rel
    top5 = np.argsort(want)[-5:]
    assert int(np.argmax(got)) in top5
    # decode a few steps through the graph-captured i8 path
    eng.decode(5)
    toks = [list(eng.gen_tokens(s)) for s in range(4)]
    assert toks[0] == toks[1] == toks[2] == toks[3]


def test_engine_mha_g1(core, tmp_path):
    """MHA layout (n_kv_heads == n_heads, GQA ratio G=1 — the llama-2
    presets) matches the numpy reference exactly (greedy tokens). Verified
    on hardware via scripts/g1_check.py before landing."""
    import numpy as np
    from crowdllama_amd.engine.ref_numpy import RefLlama
    from crowdllama_amd.models.presets import ModelConfig
    from crowdllama_amd.models.synth import write_synthetic_gguf

    cfg = ModelConfig("mha", vocab_size=256, hidden_size=256, n_layers=2,
                      n_heads=4, n_kv_heads=4, ffn_hidden=512,
                      rope_theta=10000.0, max_seq_len=256)
    path = str(tmp_path / "mha.gguf")
    write_synthetic_gguf(path, cfg, scheme="q4_k_m", mode="exact", seed=3)
    ec = core.EngineConfig()
    ec.batch = 1
    ec.max_seq = 128
    ec.act_q8 = False  # exact-greedy vs the f32 numpy reference
    eng = core.Engine(path, ec)
    want = RefLlama(path).generate([5, 9, 2, 7], 8)
    eng.prefill(np.asarray([[5, 9, 2, 7]], dtype=np.int32))
    eng.decode(7)
    assert list(eng.gen_tokens(0)) == want
