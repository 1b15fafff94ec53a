import sys, os; sys.path.insert(0, "/root/repo")
import numpy as np
from crowdllama_amd.models.presets import ModelConfig
from crowdllama_amd.models.synth import write_synthetic_gguf
from crowdllama_amd.engine.ref_numpy import RefLlama
from crowdllama_amd.ops import get_core

cfg = ModelConfig("mha", vocab_size=256, hidden_size=256, n_layers=2,
                  n_heads=4, n_kv_heads=4, ffn_hidden=512,
                  rope_theta=10000.0, max_seq_len=256)
path = "/tmp/mha.gguf"
write_synthetic_gguf(path, cfg, scheme="q4_k_m", mode="exact", seed=3)
core = get_core()
ec = core.EngineConfig(); ec.batch = 1; ec.max_seq = 128
eng = core.Engine(path, ec)
ref = RefLlama(path)
prompt = [5, 9, 2, 7]
want = ref.generate(prompt, 8)
eng.prefill(np.asarray([prompt], dtype=np.int32))
eng.decode(7)
got = list(eng.gen_tokens(0))
print("ref:", want)
print("gpu:", got)
assert got == want, "G=1 MHA mismatch"
print("G=1 attention OK")
