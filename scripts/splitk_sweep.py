"""Sweep CLA_SPLITK_TARGET (split-K workgroup target for the decode GEMM)
at the bench default operating point. Run on a GPU box:

    python scripts/splitk_sweep.py
"""
import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

for tgt in (512, 768, 1024, 1536, 2560):
    env = dict(os.environ, CLA_SPLITK_TARGET=str(tgt))
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"),
         "--steps", "32", "--warmup", "8"],
        env=env, capture_output=True, text=True)
    line = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
    if not line:
        print(tgt, "FAILED", out.stdout[-200:], out.stderr[-200:])
        continue
    d = json.loads(line[-1])
    print(f"target={tgt:5d}  {d['value']:8.1f} tok/s  "
          f"{d['ms_per_step']:.3f} ms/step", flush=True)
