"""Loading/build glue for the in-tree gfx950 HIP extension.

The extension is compiled by ``build_hip.py`` (driven from
``__graft_entry__.build()`` or ``python -m mpi_operator_amd.ops.build_hip``)
into ``mpi_operator_amd/ops/_mpi_amd_hip.so`` so that the built artifact
travels with the source tree to GPU boxes.

Contract: on a machine with a GPU (``torch.cuda.is_available()``), every op in
``mpi_operator_amd.ops`` REQUIRES the extension — a missing/unbuilt extension
raises instead of silently falling back to eager PyTorch. On CPU-only
machines the ops use their PyTorch reference implementations (used by the
numerics tests as the golden model).
"""
from __future__ import annotations

import importlib
import os
import sys

_ext = None
_ext_err: Exception | None = None


def _try_load():
    global _ext, _ext_err
    if _ext is not None or _ext_err is not None:
        return
    try:
        import torch  # noqa: F401  (the .so links against libtorch)

        _ext = importlib.import_module("mpi_operator_amd.ops._mpi_amd_hip")
    except Exception as e:  # pragma: no cover - exercised on GPU boxes only
        _ext_err = e


def hip_ext():
    """Return the loaded HIP extension module, or raise loudly.

    Raising (rather than falling back) is deliberate: on a GPU box a silent
    eager fallback would invalidate every benchmark and test that claims to
    exercise the hand-written CDNA4 kernels.
    """
    _try_load()
    if _ext is None:
        raise RuntimeError(
            "mpi_operator_amd HIP extension (_mpi_amd_hip.so) is not available "
            "but a GPU op was requested. Build it in-tree with "
            "`python -m mpi_operator_amd.ops.build_hip` (hipcc, gfx950). "
            f"Original import error: {_ext_err!r}"
        )
    return _ext


def hip_available() -> bool:
    _try_load()
    return _ext is not None
