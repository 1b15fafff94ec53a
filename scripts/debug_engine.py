"""Bisect engine-vs-reference numerics stage by stage (GPU box)."""

import sys

import numpy as np

sys.path.insert(0, ".")

from crowdllama_amd.models.presets import ModelConfig          # noqa: E402
from crowdllama_amd.models.synth import write_synthetic_gguf   # noqa: E402
from crowdllama_amd.engine.ref_numpy import RefLlama           # noqa: E402
from crowdllama_amd.ops import get_core                        # noqa: E402


def check(name, cfg, scheme, prompt):
    core = get_core()
    path = f"/tmp/dbg_{name}.gguf"
    write_synthetic_gguf(path, cfg, scheme=scheme, mode="exact", seed=11)
    ec = core.EngineConfig()
    ec.batch = 1
    ec.max_seq = 64
    eng = core.Engine(path, ec)
    eng.prefill(np.array([prompt], dtype=np.int32))
    got = np.asarray(eng.logits(0))
    ref = RefLlama(path)
    want = None
    for t in prompt:
        want = ref.step(t)
    rel = np.abs(got - want).max() / (np.abs(want).max() + 1e-9)
    print(f"{name:24s} scheme={scheme:6s} len={len(prompt)} "
          f"maxrel={rel:.3e} argmax {int(np.argmax(got))} vs {int(np.argmax(want))}")
    return rel


base = dict(vocab_size=512, hidden_size=256, n_heads=4, n_kv_heads=2,
            ffn_hidden=512, rope_theta=10000.0, max_seq_len=512)

check("L0-f32-1tok", ModelConfig("dbg0", n_layers=0, **base), "f32", [3])
check("L1-f32-1tok", ModelConfig("dbg1", n_layers=1, **base), "f32", [3])
check("L1-f32-3tok", ModelConfig("dbg1b", n_layers=1, **base), "f32", [3, 17, 99])
check("L2-f32-5tok", ModelConfig("dbg2", n_layers=2, **base), "f32", [3, 17, 99, 250, 7])
check("L2-q4km-5tok", ModelConfig("dbg3", n_layers=2, **base), "q4_k_m", [3, 17, 99, 250, 7])
