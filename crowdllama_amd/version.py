"""Version info (reference parity: pkg/version/version.go:9-47).

The reference injects Version/CommitHash/BuildDate via ldflags; here the
stamp comes from CROWDLLAMA_VERSION / CROWDLLAMA_COMMIT env (set by the
Dockerfiles' build args) with a lazy git fallback, so advertised peer
metadata carries it (reference stamps CommitHash into metadata at
pkg/peer/peer.go:335).
"""

import functools
import os
import subprocess

__version__ = os.environ.get("CROWDLLAMA_VERSION", "0.1.0")


@functools.lru_cache(maxsize=1)
def commit_hash() -> str:
    env = os.environ.get("CROWDLLAMA_COMMIT")
    if env and env != "unknown":
        return env
    try:
        out = subprocess.run(
            ["git", "rev-parse", "--short", "HEAD"],
            capture_output=True, text=True, timeout=5,
            cwd=__file__.rsplit("/", 2)[0],
        )
        if out.returncode == 0:
            return out.stdout.strip()
    except Exception:
        pass
    return "unknown"


def version_string() -> str:
    return f"crowdllama-amd {__version__} ({commit_hash()})"
