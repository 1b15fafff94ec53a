"""HIP/CDNA4 engine extension loader.

On a GPU box the native extension MUST load — ops fail loudly rather than
silently falling back to an eager path (round-end checks verify the .so the
GPU processes actually loaded).
"""

from __future__ import annotations

_mod = None


def get_core(required: bool = True):
    """Return the native module, building it first if missing/stale."""
    global _mod
    if _mod is not None:
        return _mod
    from . import build as _b
    try:
        if _b.needs_build():
            _b.build(verbose=False)
    except Exception as e:
        if required:
            raise RuntimeError(
                f"crowdllama_amd native extension build failed: {e}") from e
        return None
    try:
        import importlib
        mod = importlib.import_module(__name__ + "._core")
    except ImportError as e:
        if required:
            raise RuntimeError(
                f"crowdllama_amd native extension failed to load: {e}") from e
        return None
    _mod = mod
    return _mod


def has_gpu() -> bool:
    try:
        core = get_core(required=False)
        return bool(core and core.device_count() > 0)
    except Exception:
        return False
