"""In-tree build of the gfx950 HIP extension (no hipify, no JIT cache):
explicit hipcc invocations so the built .so lives next to the sources and
travels with the repo snapshot to GPU boxes.

Usage: python -m mpi_operator_amd.ops.build_hip [--force]
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

HERE = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(HERE, "csrc")
OUT_SO = os.path.join(HERE, "_mpi_amd_hip.so")
OBJ_DIR = os.path.join(HERE, "_build")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
HIPCC = os.environ.get("HIPCC", "hipcc")

KERNEL_SRCS = ["elementwise.hip", "batchnorm.hip", "pooling.hip",
               "softmax_xent.hip", "masked_xent.hip", "attention.hip", "gemm.hip", "conv.hip", "layernorm.hip"]
BINDING_SRC = "bindings.cpp"

BASE_FLAGS = [f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
              "-D__HIP_PLATFORM_AMD__=1"]


def _torch_flags():
    import torch
    from torch.utils import cpp_extension as ce

    inc = [f"-I{p}" for p in ce.include_paths()] + [f"-I{sysconfig.get_paths()['include']}"]
    defs = [
        "-DUSE_ROCM=1", "-DHIPBLAS_V2",
        "-DCUDA_HAS_FP16=1", "-D__HIP_NO_HALF_OPERATORS__=1",
        "-D__HIP_NO_HALF_CONVERSIONS__=1", "-DHIP_ENABLE_WARP_SYNC_BUILTINS=1",
        "-DTORCH_EXTENSION_NAME=_mpi_amd_hip",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        f"-D_GLIBCXX_USE_CXX11_ABI={int(torch._C._GLIBCXX_USE_CXX11_ABI)}",
    ]
    libs = [f"-L{p}" for p in ce.library_paths()] + [
        "-ltorch", "-ltorch_cpu", "-ltorch_python", "-lc10", "-lamdhip64",
        "-ltorch_hip", "-lc10_hip",
    ]
    return inc, defs, libs


def _stale(obj: str, src: str) -> bool:
    if not os.path.exists(obj):
        return True
    # EVERY header in csrc counts: conv.o once went stale against a
    # pipe_mix.h-only edit because this list was hand-maintained
    import glob as _glob
    dep = [src] + _glob.glob(os.path.join(CSRC, "*.h"))
    om = os.path.getmtime(obj)
    return any(os.path.getmtime(d) > om for d in dep if os.path.exists(d))


def _run(cmd):
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        sys.stderr.write(" ".join(cmd) + "\n" + r.stdout[-4000:] + r.stderr[-8000:])
        raise RuntimeError(f"build failed: {cmd[0]} {os.path.basename(cmd[-3])}")


def build(force: bool = False, verbose: bool = True) -> str:
    os.makedirs(OBJ_DIR, exist_ok=True)
    objs = []
    relink = force or not os.path.exists(OUT_SO)
    for src in KERNEL_SRCS:
        sp = os.path.join(CSRC, src)
        op = os.path.join(OBJ_DIR, src.replace(".hip", ".o"))
        objs.append(op)
        if force or _stale(op, sp):
            if verbose:
                print(f"[build_hip] hipcc -c {src}", flush=True)
            _run([HIPCC, *BASE_FLAGS, "-x", "hip", "-c", sp, "-o", op])
            relink = True
    inc, defs, libs = _torch_flags()
    bp = os.path.join(CSRC, BINDING_SRC)
    bo = os.path.join(OBJ_DIR, "bindings.o")
    objs.append(bo)
    if force or _stale(bo, bp):
        if verbose:
            print("[build_hip] hipcc -c bindings.cpp (torch headers — slow)", flush=True)
        _run([HIPCC, *BASE_FLAGS, *inc, *defs, "-x", "hip", "-c", bp, "-o", bo])
        relink = True
    if relink:
        if verbose:
            print("[build_hip] linking _mpi_amd_hip.so", flush=True)
        _run([HIPCC, "-shared", "-fPIC", *objs, *libs, "-o", OUT_SO])
    return OUT_SO


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(OUT_SO)
