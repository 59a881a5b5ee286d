"""Build/version info — the reference injects Version/GitSHA/Built via
ldflags (its pkg/version); here it is resolved at import from git when
available, else the packaged constants."""
from __future__ import annotations

import os
import subprocess

__version__ = "0.1.0"


def git_sha() -> str:
    try:
        root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        return subprocess.run(["git", "rev-parse", "--short", "HEAD"], cwd=root,
                              capture_output=True, text=True, timeout=5).stdout.strip() or "unknown"
    except Exception:
        return "unknown"


def version_string() -> str:
    return f"mpi-operator-amd {__version__} (git {git_sha()})"
