"""Synthetic random-init GGUF checkpoint generation.

There is no network access for real checkpoints (BASELINE.json: "synthetic
prompts / random-init GGUF weights"), so benches and tests generate
random-init GGUF files of the real architectures.

Two modes:
- "fast": tensor payloads are random quantized *bytes* with calibrated
  constant block scales (so dequantized weights have ~N(0, 1/fan_in)
  statistics). Generates an 8B-class Q4_K_M checkpoint in seconds —
  identical compute/memory behavior to a real checkpoint.
- "exact": quantizes gaussian floats through the reference quantizers
  (slow; used for small models in numerics tests).

Tensor naming and quant-type assignment follow the GGUF llama convention
(Q4_K_M: matrices Q4_K, output head Q6_K, norms F32).
"""

from __future__ import annotations

import hashlib
import os
import tempfile

import numpy as np

from ..quant.gguf import GGUFWriter
from ..quant.kquants import (
    GGMLType, QK_K, Q4_K_BLOCK_BYTES, Q6_K_BLOCK_BYTES, Q8_0_BLOCK,
    Q8_0_BLOCK_BYTES, dequantize, quantize,
)
from .presets import ModelConfig, get_preset

QUANT_SCHEMES = {
    # scheme -> (matrix type, output-head type, embed type)
    "q4_k_m": (GGMLType.Q4_K, GGMLType.Q6_K, GGMLType.Q4_K),
    "q8_0": (GGMLType.Q8_0, GGMLType.Q8_0, GGMLType.Q8_0),
    "q6_k": (GGMLType.Q6_K, GGMLType.Q6_K, GGMLType.Q6_K),
    "bf16": (GGMLType.BF16, GGMLType.BF16, GGMLType.BF16),
    "f16": (GGMLType.F16, GGMLType.F16, GGMLType.F16),
    "f32": (GGMLType.F32, GGMLType.F32, GGMLType.F32),
}


def _calibrate_unit_std(t: GGMLType, rng: np.random.Generator) -> float:
    """std of a dequantized fast-mode block with unit scale field(s)."""
    raw = _random_blocks(t, 64 * 1024, rng, d=1.0)
    vals = dequantize(raw.reshape(1, -1), t, 64 * 1024)
    return float(np.std(vals))


def _random_blocks(t: GGMLType, n_elems: int, rng: np.random.Generator,
                   d: float) -> np.ndarray:
    """Random quantized bytes for n_elems weights with constant scale d."""
    if t == GGMLType.Q4_K:
        nb = n_elems // QK_K
        out = np.frombuffer(rng.bytes(nb * Q4_K_BLOCK_BYTES),
                            dtype=np.uint8).reshape(nb, -1).copy()
        d16 = np.float16(d); dmin16 = np.float16(d * 7.5)
        out[:, 0:2] = np.frombuffer(d16.tobytes(), dtype=np.uint8)
        out[:, 2:4] = np.frombuffer(dmin16.tobytes(), dtype=np.uint8)
        return out.reshape(-1)
    if t == GGMLType.Q6_K:
        nb = n_elems // QK_K
        out = np.frombuffer(rng.bytes(nb * Q6_K_BLOCK_BYTES),
                            dtype=np.uint8).reshape(nb, -1).copy()
        d16 = np.float16(d)
        out[:, 208:210] = np.frombuffer(d16.tobytes(), dtype=np.uint8)
        return out.reshape(-1)
    if t == GGMLType.Q8_0:
        nb = n_elems // Q8_0_BLOCK
        out = np.frombuffer(rng.bytes(nb * Q8_0_BLOCK_BYTES),
                            dtype=np.uint8).reshape(nb, -1).copy()
        d16 = np.float16(d)
        out[:, 0:2] = np.frombuffer(d16.tobytes(), dtype=np.uint8)
        return out.reshape(-1)
    if t == GGMLType.BF16:
        # random mantissa+sign with a fixed exponent chosen so the value
        # magnitude ~ d (fast: no float RNG for 100GB-class tensors)
        e = int(np.clip(127 + np.floor(np.log2(max(d, 1e-30)) + 0.5), 1, 254))
        r = np.frombuffer(rng.bytes(n_elems * 2), dtype=np.uint16)
        bits = (r & 0x8000) | (np.uint16(e) << 7) | (r & 0x7F)
        return bits.view(np.uint8)
    if t == GGMLType.F16:
        e16 = int(np.clip(15 + np.floor(np.log2(max(d, 1e-30)) + 0.5), 1, 30))
        r = np.frombuffer(rng.bytes(n_elems * 2), dtype=np.uint16)
        bits = (r & 0x8000) | (np.uint16(e16) << 10) | (r & 0x3FF)
        return bits.view(np.uint8)
    if t == GGMLType.F32:
        vals = rng.standard_normal(n_elems, dtype=np.float32) * d
        return quantize(vals, t).reshape(-1)
    raise ValueError(f"fast mode unsupported for {t}")


class _TensorGen:
    def __init__(self, mode: str, seed: int):
        self.mode = mode
        self.rng = np.random.default_rng(seed)
        self._unit_std: dict[GGMLType, float] = {}

    def make(self, shape: tuple[int, ...], t: GGMLType,
             target_std: float) -> np.ndarray:
        n = int(np.prod(shape, dtype=np.int64))
        if self.mode == "fast":
            if t not in self._unit_std:
                self._unit_std[t] = _calibrate_unit_std(t, np.random.default_rng(0))
            d = target_std / self._unit_std[t]
            return _random_blocks(t, n, self.rng, d)
        vals = self.rng.standard_normal(n, dtype=np.float32) * target_std
        return quantize(vals.reshape(shape), t).reshape(-1)


def write_synthetic_gguf(path: str, model: str | ModelConfig,
                         scheme: str = "q4_k_m", mode: str = "fast",
                         seed: int = 1234) -> str:
    cfg = model if isinstance(model, ModelConfig) else get_preset(model)
    mat_t, head_t, embed_t = QUANT_SCHEMES[scheme]
    gen = _TensorGen(mode, seed)
    w = GGUFWriter(path)
    w.add("general.architecture", "llama")
    w.add("general.name", cfg.name)
    w.add("general.file_type", {"q4_k_m": 15, "q8_0": 7, "q6_k": 18,
                                "f16": 1, "bf16": 32, "f32": 0}[scheme])
    w.add("llama.block_count", cfg.n_layers)
    w.add("llama.context_length", cfg.max_seq_len)
    w.add("llama.embedding_length", cfg.hidden_size)
    w.add("llama.feed_forward_length", cfg.ffn_hidden)
    w.add("llama.attention.head_count", cfg.n_heads)
    w.add("llama.attention.head_count_kv", cfg.n_kv_heads)
    w.add("llama.attention.layer_norm_rms_epsilon", cfg.rms_eps)
    w.add("llama.rope.freq_base", cfg.rope_theta)
    w.add("llama.vocab_size", cfg.vocab_size)
    w.add("llama.rope.dimension_count", cfg.head_dim)
    # minimal byte-ish vocab so the tokenizer round-trips
    w.add("tokenizer.ggml.model", "gpt2")
    tokens = ["<unk>", "<s>", "</s>"] + [f"<0x{i:02X}>" for i in range(256)]
    tokens += [f"tok{i}" for i in range(cfg.vocab_size - len(tokens))]
    w.add("tokenizer.ggml.tokens", tokens)
    w.add("tokenizer.ggml.bos_token_id", 1)
    w.add("tokenizer.ggml.eos_token_id", 2)

    h, f, v = cfg.hidden_size, cfg.ffn_hidden, cfg.vocab_size
    kv_dim = cfg.n_kv_heads * cfg.head_dim
    sd_h = 1.0 / np.sqrt(h)
    sd_f = 1.0 / np.sqrt(f)

    def tensor(name, shape, t, std):
        w.add_tensor(name, shape, t, gen.make(shape, t, std))

    tensor("token_embd.weight", (v, h), embed_t, 1.0)
    for i in range(cfg.n_layers):
        p = f"blk.{i}."
        w.add_tensor(p + "attn_norm.weight", (h,), GGMLType.F32,
                     np.ones(h, dtype=np.float32).view(np.uint8))
        tensor(p + "attn_q.weight", (h, h), mat_t, sd_h)
        tensor(p + "attn_k.weight", (kv_dim, h), mat_t, sd_h)
        tensor(p + "attn_v.weight", (kv_dim, h), mat_t, sd_h)
        tensor(p + "attn_output.weight", (h, h), mat_t, sd_h)
        w.add_tensor(p + "ffn_norm.weight", (h,), GGMLType.F32,
                     np.ones(h, dtype=np.float32).view(np.uint8))
        tensor(p + "ffn_gate.weight", (f, h), mat_t, sd_h)
        tensor(p + "ffn_up.weight", (f, h), mat_t, sd_h)
        tensor(p + "ffn_down.weight", (h, f), mat_t, sd_f)
    w.add_tensor("output_norm.weight", (h,), GGMLType.F32,
                 np.ones(h, dtype=np.float32).view(np.uint8))
    tensor("output.weight", (v, h), head_t, sd_h)
    w.write()
    return path


def synth_path(model: str, scheme: str = "q4_k_m", mode: str = "fast",
               seed: int = 1234, cache_dir: str | None = None) -> str:
    """Cached path for a synthetic checkpoint (generate if absent)."""
    cache_dir = cache_dir or os.path.join(tempfile.gettempdir(),
                                          "crowdllama_amd_models")
    os.makedirs(cache_dir, exist_ok=True)
    key = f"{model}-{scheme}-{mode}-{seed}"
    tag = hashlib.sha1(key.encode()).hexdigest()[:10]
    path = os.path.join(cache_dir, f"{model}-{scheme}-{tag}.gguf")
    if not os.path.exists(path):
        tmp = path + f".tmp.{os.getpid()}"
        write_synthetic_gguf(tmp, model, scheme, mode, seed)
        os.replace(tmp, path)
    return path
