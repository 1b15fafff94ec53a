"""In-tree build of the HIP engine extension for gfx950.

Invoked by setup.py, __graft_entry__.build(), and lazily on first import
attempt. hipcc cross-compiles on CPU-only boxes; the resulting .so travels
with the repo snapshot to GPU boxes (it is git-ignored but NOT
gpurun-ignored).
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

OPS_DIR = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(OPS_DIR, "csrc")
ARCH = os.environ.get("CLA_GFX_ARCH", "gfx950")

SOURCES = ["gguf.cpp", "engine.cpp", "testutil.cpp", "bindings.cpp",
           "kernels.hip", "mfma_probe.hip", "gemm.hip", "gemm_i8.hip",
           "tokenizer.cpp"]


def _ext_suffix() -> str:
    return sysconfig.get_config_var("EXT_SUFFIX") or ".so"


def so_path() -> str:
    return os.path.join(OPS_DIR, "_core" + _ext_suffix())


def _newest_src_mtime() -> float:
    return max(os.path.getmtime(os.path.join(CSRC, s)) for s in SOURCES)


def needs_build() -> bool:
    so = so_path()
    return (not os.path.exists(so)) or os.path.getmtime(so) < _newest_src_mtime()


def build(verbose: bool = True, force: bool = False) -> str:
    so = so_path()
    if not force and not needs_build():
        return so
    import pybind11
    hipcc = os.environ.get("HIPCC", "hipcc")
    py_inc = sysconfig.get_paths()["include"]
    objs = []
    os.makedirs(os.path.join(OPS_DIR, "build"), exist_ok=True)
    procs = []
    for src in SOURCES:
        obj = os.path.join(OPS_DIR, "build", src.rsplit(".", 1)[0] + ".o")
        objs.append(obj)
        cmd = [hipcc, f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
               *(["-DCLA_PERF_PROBE_NOATOMIC"] if os.environ.get("CLA_PROBE") else []),
               "-I", CSRC, "-I", py_inc, "-I", pybind11.get_include(),
               "-c", os.path.join(CSRC, src), "-o", obj]
        if src.endswith(".hip"):
            cmd.insert(1, "-x")
            cmd.insert(2, "hip")
        if verbose:
            print("[ops.build]", " ".join(cmd), file=sys.stderr)
        procs.append(subprocess.Popen(cmd, stdout=subprocess.PIPE,
                                      stderr=subprocess.PIPE, text=True))
    for p, src in zip(procs, SOURCES):
        out, err = p.communicate()
        if p.returncode != 0:
            raise RuntimeError(f"hipcc failed on {src}:\n{out}\n{err}")
        if verbose and err.strip():
            print(err, file=sys.stderr)
    link = [hipcc, f"--offload-arch={ARCH}", "-shared", "-fPIC", *objs,
            "-L/opt/rocm/lib", "-lrccl", "-o", so]
    if verbose:
        print("[ops.build]", " ".join(link), file=sys.stderr)
    r = subprocess.run(link, capture_output=True, text=True)
    if r.returncode != 0:
        raise RuntimeError(f"hipcc link failed:\n{r.stdout}\n{r.stderr}")
    return so


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(so_path())
