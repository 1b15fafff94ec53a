"""Model-family presets (llama architecture family).

The reference advertises/serves these model names (hardcoded list at
reference pkg/peer/peer.go:322 and the BASELINE.json configs); here they map
to real architecture hyperparameters used for synthetic random-init
checkpoints and engine graph construction.
"""

from __future__ import annotations

from dataclasses import dataclass


@dataclass(frozen=True)
class ModelConfig:
    name: str
    vocab_size: int
    hidden_size: int
    n_layers: int
    n_heads: int
    n_kv_heads: int
    ffn_hidden: int
    rope_theta: float = 10000.0
    rms_eps: float = 1e-5
    max_seq_len: int = 8192
    tie_embeddings: bool = False

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.n_heads

    def n_params(self) -> int:
        h, v, f = self.hidden_size, self.vocab_size, self.ffn_hidden
        kvh = self.n_kv_heads * self.head_dim
        per_layer = (h * h + 2 * h * kvh + h * h      # q, k, v, o
                     + 3 * h * f                       # gate, up, down
                     + 2 * h)                          # norms
        embed = v * h * (1 if self.tie_embeddings else 2)
        return self.n_layers * per_layer + embed + h


PRESETS: dict[str, ModelConfig] = {}


def _reg(cfg: ModelConfig) -> ModelConfig:
    PRESETS[cfg.name] = cfg
    return cfg


# BASELINE.json config models
_reg(ModelConfig("tinyllama", vocab_size=32000, hidden_size=2048, n_layers=22,
                 n_heads=32, n_kv_heads=4, ffn_hidden=5632,
                 rope_theta=10000.0, max_seq_len=2048))
_reg(ModelConfig("llama3-8b", vocab_size=128256, hidden_size=4096, n_layers=32,
                 n_heads=32, n_kv_heads=8, ffn_hidden=14336,
                 rope_theta=500000.0, max_seq_len=8192))
_reg(ModelConfig("mistral-7b", vocab_size=32000, hidden_size=4096, n_layers=32,
                 n_heads=32, n_kv_heads=8, ffn_hidden=14336,
                 rope_theta=10000.0, max_seq_len=8192))
_reg(ModelConfig("llama3-70b", vocab_size=128256, hidden_size=8192, n_layers=80,
                 n_heads=64, n_kv_heads=8, ffn_hidden=28672,
                 rope_theta=500000.0, max_seq_len=8192))
# the reference's hardcoded advertised list also names llama-2 models
# (peer.go:322); MHA (n_kv_heads == n_heads, G=1) is a supported layout
_reg(ModelConfig("llama-2-7b", vocab_size=32000, hidden_size=4096,
                 n_layers=32, n_heads=32, n_kv_heads=32, ffn_hidden=11008,
                 rope_theta=10000.0, max_seq_len=4096))
_reg(ModelConfig("llama-2-13b", vocab_size=32000, hidden_size=5120,
                 n_layers=40, n_heads=40, n_kv_heads=40, ffn_hidden=13824,
                 rope_theta=10000.0, max_seq_len=4096))
# tiny model for tests (fast CPU generation + load)
_reg(ModelConfig("testllama", vocab_size=512, hidden_size=256, n_layers=2,
                 n_heads=4, n_kv_heads=2, ffn_hidden=512,
                 rope_theta=10000.0, max_seq_len=512))

# aliases matching Ollama-style names used by the reference / baseline
ALIASES = {
    "llama3:8b": "llama3-8b",
    "llama3:70b": "llama3-70b",
    "mistral:7b": "mistral-7b",
}


def get_preset(name: str) -> ModelConfig:
    name = name.strip()
    if name in PRESETS:
        return PRESETS[name]
    if name in ALIASES:
        return PRESETS[ALIASES[name]]
    raise KeyError(f"unknown model preset: {name!r} "
                   f"(known: {sorted(PRESETS) + sorted(ALIASES)})")
