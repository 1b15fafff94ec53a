"""Bench the model/scheme matrix on one GPU (BASELINE configs)."""
import json
import subprocess
import sys

CASES = [
    ("tinyllama", "q8_0", 1, 128),     # BASELINE config 1 model (on GPU)
    ("llama3-8b", "q4_k_m", 1, 256),   # headline
    ("mistral-7b", "q4_k_m", 1, 256),
    ("llama3-8b", "q8_0", 1, 128),
]
for model, scheme, batch, steps in CASES:
    r = subprocess.run(
        [sys.executable, "bench.py", "--gpus", "1", "--steps", str(steps),
         "--warmup", "16", "--model", model, "--scheme", scheme,
         "--batch", str(batch)],
        capture_output=True, text=True, timeout=900)
    line = [ln for ln in r.stdout.splitlines() if ln.startswith("{")]
    if not line:
        print(f"{model:12s} {scheme:7s} FAILED: {r.stderr[-200:]}")
        continue
    d = json.loads(line[-1])
    print(f"{model:12s} {scheme:7s} B={batch}: {d['value']:8.1f} tok/s  "
          f"{d['ms_per_step']:.2f} ms/step  prefill {d['config']['prefill_ms']:.0f} ms")
