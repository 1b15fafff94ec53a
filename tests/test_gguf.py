"""GGUF v3 container round-trip + synthetic checkpoint tests."""

import numpy as np
import pytest

from crowdllama_amd.models import get_preset, write_synthetic_gguf
from crowdllama_amd.quant import GGMLType, GGUFReader, GGUFWriter, quantize


def test_gguf_roundtrip(tmp_path):
    path = str(tmp_path / "t.gguf")
    w = GGUFWriter(path)
    w.add("general.architecture", "llama")
    w.add("test.int", 42)
    w.add("test.float", 2.5)
    w.add("test.bool", True)
    w.add("test.string", "hello world")
    w.add("test.array", [1, 2, 3])
    w.add("test.strarray", ["a", "bc", "def"])
    rng = np.random.default_rng(0)
    a = rng.standard_normal((16, 256)).astype(np.float32)
    b = rng.standard_normal((8, 512)).astype(np.float32)
    w.add_tensor("a", a.shape, GGMLType.F32, a.view(np.uint8))
    w.add_tensor("b.q4k", b.shape, GGMLType.Q4_K, quantize(b, GGMLType.Q4_K))
    w.write()

    with GGUFReader(path) as r:
        assert r.metadata["general.architecture"] == "llama"
        assert r.metadata["test.int"] == 42
        assert abs(r.metadata["test.float"] - 2.5) < 1e-9
        assert r.metadata["test.bool"] is True
        assert r.metadata["test.string"] == "hello world"
        assert r.metadata["test.array"] == [1, 2, 3]
        assert r.metadata["test.strarray"] == ["a", "bc", "def"]
        assert r.tensors["a"].shape == (16, 256)
        assert r.tensors["a"].ggml_type == GGMLType.F32
        np.testing.assert_array_equal(r.tensor_f32("a"), a)
        y = r.tensor_f32("b.q4k")
        assert y.shape == (8, 512)
        assert np.corrcoef(y.ravel(), b.ravel())[0, 1] > 0.99


@pytest.mark.parametrize("scheme", ["q4_k_m", "q8_0"])
def test_synthetic_model(tmp_path, scheme):
    cfg = get_preset("testllama")
    path = str(tmp_path / "m.gguf")
    write_synthetic_gguf(path, "testllama", scheme=scheme, mode="fast")
    with GGUFReader(path) as r:
        assert r.metadata["general.architecture"] == "llama"
        assert r.metadata["llama.block_count"] == cfg.n_layers
        assert r.metadata["llama.embedding_length"] == cfg.hidden_size
        names = set(r.tensors)
        assert "token_embd.weight" in names
        assert "output.weight" in names
        assert "blk.0.attn_q.weight" in names
        assert "blk.1.ffn_down.weight" in names
        # dequantized stats: roughly unit-free scale, no NaN/inf
        qw = r.tensor_f32("blk.0.attn_q.weight")
        assert qw.shape == (cfg.hidden_size, cfg.hidden_size)
        assert np.isfinite(qw).all()
        std = qw.std()
        target = 1.0 / np.sqrt(cfg.hidden_size)
        assert 0.5 * target < std < 2.0 * target
        emb = r.tensor_f32("token_embd.weight")
        assert np.isfinite(emb).all()
        assert 0.5 < emb.std() < 2.0


def test_synthetic_exact_mode(tmp_path):
    path = str(tmp_path / "m.gguf")
    write_synthetic_gguf(path, "testllama", scheme="q8_0", mode="exact")
    with GGUFReader(path) as r:
        qw = r.tensor_f32("blk.0.attn_q.weight")
        assert np.isfinite(qw).all()


def test_reader_rejects_bad_magic(tmp_path):
    p = tmp_path / "bad.gguf"
    p.write_bytes(b"NOPE" + b"\x00" * 64)
    with pytest.raises(Exception):
        GGUFReader(str(p))


def test_reader_rejects_truncated(tmp_path):
    path = str(tmp_path / "t.gguf")
    w = GGUFWriter(path)
    w.add("general.architecture", "llama")
    w.add_tensor("a", (4, 32), GGMLType.F32,
                 quantize(np.zeros((4, 32), dtype=np.float32),
                          GGMLType.F32).reshape(4, -1))
    w.write()
    data = open(path, "rb").read()
    trunc = str(tmp_path / "trunc.gguf")
    open(trunc, "wb").write(data[: len(data) // 2])
    with pytest.raises(Exception):
        r = GGUFReader(trunc)
        # if the header parsed, reading the tensor must still fail
        r.tensor_data("a")


def test_reader_unknown_tensor_errors(tmp_path):
    path = str(tmp_path / "t.gguf")
    w = GGUFWriter(path)
    w.add("general.architecture", "llama")
    w.add_tensor("a", (2, 32), GGMLType.F32,
                 quantize(np.zeros((2, 32), dtype=np.float32),
                          GGMLType.F32).reshape(2, -1))
    w.write()
    with GGUFReader(path) as r:
        with pytest.raises(KeyError):
            r.tensor_data("nope")
