"""Host-side sampling (temperature / top-k / top-p) over engine logits.

The hot serving path uses the engine's on-device greedy argmax; when a
request asks for temperature sampling the worker switches to this
logits-copy-back path (reference parity: Ollama's sampling options carried
through /api/chat options)."""

from __future__ import annotations

import numpy as np


def sample(logits: np.ndarray, temperature: float = 0.0, top_k: int = 0,
           top_p: float = 0.0, rng: np.random.Generator | None = None) -> int:
    logits = np.asarray(logits, dtype=np.float32).reshape(-1)
    if temperature <= 0.0:
        return int(np.argmax(logits))
    rng = rng or np.random.default_rng()
    x = logits / max(temperature, 1e-6)
    if top_k and 0 < top_k < x.size:
        kth = np.partition(x, -top_k)[-top_k]
        x = np.where(x < kth, -np.inf, x)
    x = x - x.max()
    p = np.exp(x)
    p /= p.sum()
    if top_p and 0.0 < top_p < 1.0:
        order = np.argsort(-p)
        csum = np.cumsum(p[order])
        cut = int(np.searchsorted(csum, top_p)) + 1
        mask = np.zeros_like(p)
        mask[order[:cut]] = p[order[:cut]]
        p = mask / mask.sum()
    return int(rng.choice(p.size, p=p))
