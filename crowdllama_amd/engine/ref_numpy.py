"""Pure-numpy fp32 reference implementation of the llama decode path.

Numerics ground truth for the HIP engine: loads the same GGUF checkpoint,
mirrors the engine's exact semantics (NORM-style RoPE over adjacent pairs,
bf16-rounded KV cache, greedy argmax), and is compared against the GPU
engine in tests/test_gpu_engine.py. Deliberately slow and simple.
"""

from __future__ import annotations

import numpy as np

from ..quant.gguf import GGUFReader


def _bf16_round(x: np.ndarray) -> np.ndarray:
    u = x.astype(np.float32).view(np.uint32)
    r = ((u + 0x7FFF + ((u >> 16) & 1)) & 0xFFFF0000).view(np.float32)
    return r


def _softmax(x: np.ndarray, axis: int = -1) -> np.ndarray:
    m = x.max(axis=axis, keepdims=True)
    e = np.exp(x - m)
    return e / e.sum(axis=axis, keepdims=True)


def _act_q8(x: np.ndarray) -> np.ndarray:
    """Per-32 symmetric int8 activation quantization — exactly the engine's
    k_gemv_q8 staging (xd = amax/127, rint round-half-even)."""
    b = x.reshape(-1, 32).astype(np.float32)
    amax = np.abs(b).max(axis=1, keepdims=True)
    rinv = np.where(amax > 0, 127.0 / np.where(amax == 0, 1, amax), 0.0)
    q = np.rint(b * rinv)
    xd = amax / 127.0
    return (q * xd).reshape(x.shape).astype(np.float32)


class RefLlama:
    def __init__(self, gguf_path: str, kv_bf16: bool = True,
                 act_q8: bool = False):
        self.act_q8 = act_q8
        self.r = GGUFReader(gguf_path)
        md = self.r.metadata
        self.n_layers = md["llama.block_count"]
        self.hidden = md["llama.embedding_length"]
        self.heads = md["llama.attention.head_count"]
        self.kv_heads = md.get("llama.attention.head_count_kv", self.heads)
        self.ffn = md["llama.feed_forward_length"]
        self.theta = md.get("llama.rope.freq_base", 10000.0)
        self.eps = md.get("llama.attention.layer_norm_rms_epsilon", 1e-5)
        self.head_dim = self.hidden // self.heads
        self.vocab = md.get("llama.vocab_size",
                            self.r.tensors["token_embd.weight"].shape[0])
        self.kv_bf16 = kv_bf16
        self._cache: dict[str, np.ndarray] = {}
        # KV cache: [layer][2][t][kv_heads][head_dim]
        self.kv: list[list[np.ndarray]] = [[] for _ in range(self.n_layers)]

    def w(self, name: str) -> np.ndarray:
        if name not in self._cache:
            self._cache[name] = self.r.tensor_f32(name)
        return self._cache[name]

    def _is_quant(self, name: str) -> bool:
        return int(self.r.tensors[name].ggml_type) in (8, 12, 14)

    def mm(self, name: str, x: np.ndarray) -> np.ndarray:
        """W[name] @ x with the engine's activation quantization when the
        weight is a quantized dtype and act_q8 is on."""
        if self.act_q8 and self._is_quant(name):
            x = _act_q8(x)
        return self.w(name) @ x

    def reset(self):
        self.kv = [[] for _ in range(self.n_layers)]

    def _rms(self, x: np.ndarray, g: np.ndarray) -> np.ndarray:
        inv = 1.0 / np.sqrt(np.mean(x * x) + self.eps)
        return x * inv * g

    def _rope(self, v: np.ndarray, pos: int) -> np.ndarray:
        # NORM style: adjacent pairs (2i, 2i+1), freq theta^(-2i/d)
        d = self.head_dim
        out = v.copy()
        i = np.arange(d // 2)
        ang = pos * self.theta ** (-2.0 * i / d)
        c, s = np.cos(ang), np.sin(ang)
        x0 = v[..., 0::2]
        x1 = v[..., 1::2]
        out[..., 0::2] = x0 * c - x1 * s
        out[..., 1::2] = x0 * s + x1 * c
        return out

    def step(self, token: int) -> np.ndarray:
        """Feed one token; returns logits (f32 [vocab])."""
        h, nh, nkv, hd = self.hidden, self.heads, self.kv_heads, self.head_dim
        G = nh // nkv
        x = self.w("token_embd.weight")[token].astype(np.float32)
        for li in range(self.n_layers):
            p = f"blk.{li}."
            xn = self._rms(x, self.w(p + "attn_norm.weight"))
            q = self.mm(p + "attn_q.weight", xn)
            k = self.mm(p + "attn_k.weight", xn)
            v = self.mm(p + "attn_v.weight", xn)
            pos = len(self.kv[li])
            q = self._rope(q.reshape(nh, hd), pos)
            k = self._rope(k.reshape(nkv, hd), pos)
            v = v.reshape(nkv, hd)
            if self.kv_bf16:
                k, v = _bf16_round(k), _bf16_round(v)
            self.kv[li].append((k, v))
            ks = np.stack([e[0] for e in self.kv[li]])  # [t, nkv, hd]
            vs = np.stack([e[1] for e in self.kv[li]])
            attn = np.zeros((nh, hd), dtype=np.float32)
            for head in range(nh):
                kvh = head // G
                scores = ks[:, kvh] @ q[head] / np.sqrt(hd)
                w = _softmax(scores)
                attn[head] = w @ vs[:, kvh]
            x = x + self.mm(p + "attn_output.weight", attn.reshape(-1))
            xn = self._rms(x, self.w(p + "ffn_norm.weight"))
            g = self.mm(p + "ffn_gate.weight", xn)
            u = self.mm(p + "ffn_up.weight", xn)
            act = (g / (1.0 + np.exp(-g))) * u
            x = x + self.mm(p + "ffn_down.weight", act)
        xn = self._rms(x, self.w("output_norm.weight"))
        head_name = ("output.weight" if "output.weight" in self.r.tensors
                     else "token_embd.weight")
        return self.mm(head_name, xn)

    def generate(self, prompt: list[int], n_new: int) -> list[int]:
        """Greedy generation; returns n_new tokens (incl. first post-prompt)."""
        self.reset()
        logits = None
        for t in prompt:
            logits = self.step(t)
        out = []
        cur = int(np.argmax(logits))
        out.append(cur)
        for _ in range(n_new - 1):
            logits = self.step(cur)
            cur = int(np.argmax(logits))
            out.append(cur)
        return out
