"""Tensor-parallel sharding plan for the llama family over RCCL/xGMI.

Capability extension over the reference (which has no collectives at all —
SURVEY.md §2.3: "Collectives / NCCL call sites: none exist"): one worker
spanning N GPUs with row/column-sharded projections and RCCL all-reduce
after attn-out and ffn-down, sized for 8x MI355X with 7 point-to-point xGMI
links per GPU.

The plan is pure metadata (computed here, CPU-testable); the C++ engine
consumes it at weight-upload time.
"""

from __future__ import annotations

from dataclasses import dataclass

from ..models.presets import ModelConfig


@dataclass(frozen=True)
class Shard:
    """Slice of a 2D weight [rows, cols]: rows [r0,r1), cols [c0,c1)."""
    kind: str         # "rows" | "cols" | "replicate"
    r0: int
    r1: int
    c0: int
    c1: int


def _row_shard(n_rows: int, n_cols: int, rank: int, tp: int) -> Shard:
    per = n_rows // tp
    return Shard("rows", rank * per, (rank + 1) * per, 0, n_cols)


def _col_shard(n_rows: int, n_cols: int, rank: int, tp: int) -> Shard:
    per = n_cols // tp
    return Shard("cols", 0, n_rows, rank * per, (rank + 1) * per)


def _replicate(n_rows: int, n_cols: int) -> Shard:
    return Shard("replicate", 0, n_rows, 0, n_cols)


def validate_tp(cfg: ModelConfig, tp: int) -> None:
    if tp == 1:
        return
    if cfg.n_kv_heads % tp:
        raise ValueError(f"{cfg.name}: kv_heads {cfg.n_kv_heads} "
                         f"not divisible by tp={tp}")
    if cfg.n_heads % tp or cfg.ffn_hidden % tp or cfg.vocab_size % tp:
        raise ValueError(f"{cfg.name}: heads/ffn/vocab not divisible by tp")
    # column shards must land on 256-element superblock boundaries so
    # quantized rows slice cleanly
    if (cfg.hidden_size // tp) % 256 or (cfg.ffn_hidden // tp) % 256:
        raise ValueError(f"{cfg.name}: column shards not 256-aligned at tp={tp}")


def shard_plan(cfg: ModelConfig, rank: int, tp: int) -> dict[str, Shard]:
    """tensor-name pattern -> Shard for one rank.

    Attention is head-sharded (q rows follow the kv-group ordering so each
    rank owns whole GQA groups); o and ffn_down are column-sharded with an
    all-reduce after; gate/up row-sharded; output head vocab-row-sharded
    with a logits all-gather.
    """
    h, f, v = cfg.hidden_size, cfg.ffn_hidden, cfg.vocab_size
    kv_dim = cfg.n_kv_heads * cfg.head_dim
    if tp > 1:
        validate_tp(cfg, tp)
    plan = {
        "token_embd.weight": _replicate(v, h),
        "output_norm.weight": _replicate(1, h),
        "attn_norm.weight": _replicate(1, h),
        "ffn_norm.weight": _replicate(1, h),
        "attn_q.weight": _row_shard(h, h, rank, tp),
        "attn_k.weight": _row_shard(kv_dim, h, rank, tp),
        "attn_v.weight": _row_shard(kv_dim, h, rank, tp),
        "attn_output.weight": _col_shard(h, h, rank, tp),
        "ffn_gate.weight": _row_shard(f, h, rank, tp),
        "ffn_up.weight": _row_shard(f, h, rank, tp),
        "ffn_down.weight": _col_shard(h, f, rank, tp),
        "output.weight": _row_shard(v, h, rank, tp),
    }
    return plan


def local_meta(cfg: ModelConfig, tp: int) -> dict[str, int]:
    """Per-rank attention geometry."""
    return {
        "heads": cfg.n_heads // tp,
        "kv_heads": cfg.n_kv_heads // tp,
        "vocab_shard": cfg.vocab_size // tp,
        "ffn": cfg.ffn_hidden // tp,
    }
