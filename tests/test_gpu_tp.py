"""Tensor-parallel engine on hardware: TP=2 with one rank per GPU.

Measured fact (MI355X, RCCL 2.27.7 / ROCm 7.2): RCCL REJECTS two ranks of
one communicator on the same device — ncclCommInitRank fails with
"Duplicate GPU detected : rank 0 and rank 1 both on CUDA device"
(init.cc:1164) → ncclInvalidUsage. So TP=2 inside a 1-GPU lease is
impossible at the RCCL layer; on a single-GPU box these tests skip and the
graph-captured-collective risk is covered by test_rccl_graph_capture_1rank
below (1-rank communicator: same ncclAllReduce/ncclAllGather call sites and
hipGraph capture machinery, degenerate exchange).

Reference parity note: the reference has no collectives at all (SURVEY.md
§2.3 — its only parallelism is DP request scatter, manager.go:338); TP over
RCCL/xGMI is the MI355X-native capability extension for BASELINE config 4.
"""

import multiprocessing as mp
import os

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

PROMPTS = [[3, 17, 99, 250, 7], [5, 9, 44, 2, 250]]
DECODE_STEPS = 6


def _tp_cfg():
    from crowdllama_amd.models.presets import ModelConfig
    # dims must satisfy the engine's TP constraints: kv/heads/ffn/vocab
    # divisible by tp, hidden/tp and ffn/tp 256-aligned
    return ModelConfig("tptest", vocab_size=512, hidden_size=512,
                       n_layers=2, n_heads=4, n_kv_heads=2, ffn_hidden=1024,
                       rope_theta=10000.0, max_seq_len=256)


def _rank_main(rank, path, nccl_id, use_graph, conn):
    try:
        from crowdllama_amd.ops import get_core
        core = get_core()
        cfg = core.EngineConfig()
        cfg.batch = 2
        cfg.max_seq = 128
        cfg.device = rank % max(1, core.device_count())
        cfg.tp_rank = rank
        cfg.tp_size = 2
        cfg.nccl_id = nccl_id
        cfg.use_graph = use_graph
        eng = core.Engine(path, cfg)
        eng.prefill(np.asarray(PROMPTS, dtype=np.int32))
        logits_pf = np.asarray(eng.logits(0))
        eng.decode(DECODE_STEPS)  # graph-captures the step incl. collectives
        toks = [list(eng.gen_tokens(s)) for s in range(2)]
        logits_dec = np.asarray(eng.logits(0))
        conn.send(("ok", rank, logits_pf, logits_dec, toks))
    except Exception as e:  # noqa: BLE001 — surfaced in the parent assert
        conn.send(("error", rank, repr(e), None, None))
    finally:
        conn.close()


@pytest.fixture(scope="module")
def tp_gguf(tmp_path_factory):
    from crowdllama_amd.models.synth import write_synthetic_gguf
    path = str(tmp_path_factory.mktemp("tp") / "tp.gguf")
    write_synthetic_gguf(path, _tp_cfg(), scheme="q4_k_m", mode="exact",
                         seed=21)
    return path


def _run_tp2(tp_gguf, use_graph):
    from crowdllama_amd.ops import get_core
    core = get_core()
    if core.device_count() == 0:
        pytest.skip("no GPU")
    if core.device_count() < 2:
        pytest.skip("TP=2 needs 2 GPUs: RCCL rejects two ranks on one "
                    "device (Duplicate GPU detected, ncclInvalidUsage)")
    nccl_id = core.nccl_unique_id()
    ctx = mp.get_context("spawn")
    results = {}
    pipes, procs = [], []
    try:
        for rank in range(2):
            recv, send = ctx.Pipe(duplex=False)
            p = ctx.Process(target=_rank_main,
                            args=(rank, tp_gguf, nccl_id, use_graph, send))
            p.start()
            pipes.append(recv)
            procs.append(p)
        for recv, p in zip(pipes, procs):
            assert recv.poll(300), "TP rank timed out (collective deadlock?)"
            msg = recv.recv()
            assert msg[0] == "ok", f"rank {msg[1]} failed: {msg[2]}"
            results[msg[1]] = msg[2:]
        for p in procs:
            p.join(timeout=60)
    finally:
        # never leave a rank spinning on the leased GPU
        for p in procs:
            if p.is_alive():
                p.terminate()
                p.join(timeout=10)
            if p.is_alive():
                p.kill()
    return results


def _tp1_reference(tp_gguf):
    from crowdllama_amd.ops import get_core
    core = get_core()
    cfg = core.EngineConfig()
    cfg.batch = 2
    cfg.max_seq = 128
    eng = core.Engine(tp_gguf, cfg)
    eng.prefill(np.asarray(PROMPTS, dtype=np.int32))
    logits_pf = np.asarray(eng.logits(0))
    eng.decode(DECODE_STEPS)
    toks = [list(eng.gen_tokens(s)) for s in range(2)]
    return logits_pf, toks


def test_tp2_matches_tp1(tp_gguf):
    """TP=2 logits match TP=1 within collective-reduction tolerance and the
    two ranks agree exactly with each other (post-all-gather state is
    replicated)."""
    res = _run_tp2(tp_gguf, use_graph=True)
    logits_pf0, logits_dec0, toks0 = res[0]
    logits_pf1, logits_dec1, toks1 = res[1]
    # ranks must be bit-identical after the all-gather epilogue
    np.testing.assert_array_equal(logits_pf0, logits_pf1)
    np.testing.assert_array_equal(logits_dec0, logits_dec1)
    assert toks0 == toks1

    want_pf, want_toks = _tp1_reference(tp_gguf)
    denom = np.abs(want_pf).max() + 1e-9
    rel = np.abs(logits_pf0 - want_pf).max() / denom
    assert rel < 5e-3, f"TP=2 prefill logits diverge from TP=1: {rel}"
    # greedy chains may split on near-ties; require first token + a prefix
    for s in range(2):
        assert toks0[s][0] == want_toks[s][0]
        match = 0
        for a, b in zip(toks0[s], want_toks[s]):
            if a != b:
                break
            match += 1
        assert match >= 4, f"slot {s}: {toks0[s]} vs {want_toks[s]}"


def test_rccl_graph_capture_1rank():
    """RCCL collectives inside hipGraph capture, replayed twice — the exact
    machinery the TP decode step uses (same call sites: ncclAllReduce then
    ncclAllGather on the engine stream). Runs in a 1-GPU lease via a 1-rank
    communicator."""
    from crowdllama_amd.ops import get_core
    core = get_core()
    if core.device_count() == 0:
        pytest.skip("no GPU")
    rng = np.random.default_rng(5)
    x = rng.standard_normal(4096).astype(np.float32)
    out = np.asarray(core.test_rccl_graph_1rank(x))
    np.testing.assert_allclose(out, x, rtol=0, atol=0)


def test_tp2_eager_matches_graph(tp_gguf):
    """The hipGraph-captured TP step replays the same computation as eager
    stepping (capture of in-graph RCCL collectives — SURVEY §7.3 risk)."""
    res_g = _run_tp2(tp_gguf, use_graph=True)
    res_e = _run_tp2(tp_gguf, use_graph=False)
    np.testing.assert_array_equal(res_g[0][1], res_e[0][1])
    assert res_g[0][2] == res_e[0][2]
