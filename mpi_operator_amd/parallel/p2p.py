"""Same-node xGMI P2P verification for multi-pod jobs.

SURVEY.md §7 names the baseline-destroying failure for the 2-pods × 4-GPUs
config: RCCL ranks in different pods on ONE node silently falling back to
TCP because IPC across the pod boundary is unavailable (wrong
HSA_ENABLE_IPC_MODE_LEGACY, missing /dev/kfd, NCCL_P2P_DISABLE, shm too
small). The reference has no equivalent — its data plane is the workload
image's problem (reference README.md:115) — so this is a new, MI355X-first
guard: detect same-node rank groups and PROVE the fabric is in use with a
bandwidth canary, failing loudly instead of training at TCP speed.

``verify_p2p()`` is called from the trainer/bench once the process group is
up (world > 1, CUDA). Escape hatch: MPIAMD_ALLOW_TCP=1 downgrades the
failure to a warning (e.g. deliberate cross-node TCP smoke tests).
"""
from __future__ import annotations

import os
import time
import warnings

import torch
import torch.distributed as dist

# Effective allreduce busbw below this on a same-node pair is not xGMI
# (7 links × ~153 GB/s each; even one link ≫ this; TCP/shm loopback ≲20).
_MIN_SAME_NODE_BUSBW_GB = 25.0


def _node_id() -> str:
    """Stable per-node identity that is IDENTICAL across pods on one node
    (hostname is per-pod, so use the kernel boot_id)."""
    try:
        with open("/proc/sys/kernel/random/boot_id") as f:
            return f.read().strip()
    except OSError:
        import socket
        return socket.gethostname()


def same_node_groups(group=None) -> list[list[int]]:
    """Rank lists grouped by physical node (boot_id all-gather)."""
    world = dist.get_world_size(group)
    ids = [None] * world
    dist.all_gather_object(ids, _node_id(), group=group)
    by_node: dict[str, list[int]] = {}
    for r, nid in enumerate(ids):
        by_node.setdefault(nid, []).append(r)
    return list(by_node.values())


def static_preconditions() -> list[str]:
    """Config errors that guarantee (or strongly imply) a TCP fallback for
    same-node P2P, checkable without a collective."""
    problems = []
    if not os.path.exists("/dev/kfd"):
        problems.append("/dev/kfd not visible in this container (no GPU fabric access)")
    if os.environ.get("HSA_ENABLE_IPC_MODE_LEGACY", "0") != "0":
        problems.append(
            "HSA_ENABLE_IPC_MODE_LEGACY != 0: this host driver only supports "
            "dmabuf IPC; legacy mode breaks cross-process CUDA-tensor/RCCL IPC")
    if os.environ.get("NCCL_P2P_DISABLE") == "1":
        problems.append("NCCL_P2P_DISABLE=1 forces the TCP/shm path")
    if os.environ.get("NCCL_SHM_DISABLE") == "1" and os.environ.get("NCCL_P2P_DISABLE") == "1":
        problems.append("both P2P and SHM disabled: collectives go over sockets")
    return problems


def measure_busbw(group=None, nbytes: int = 64 << 20, iters: int = 5) -> float:
    """Allreduce bus bandwidth (GB/s) over the group: 2(n-1)/n × bytes / t."""
    world = dist.get_world_size(group)
    t = torch.empty(nbytes // 2, dtype=torch.bfloat16, device="cuda")
    for _ in range(2):  # warmup
        dist.all_reduce(t, group=group)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        dist.all_reduce(t, group=group)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    return 2 * (world - 1) / world * nbytes / dt / 1e9


def verify_p2p(group=None, min_busbw: float = _MIN_SAME_NODE_BUSBW_GB) -> dict:
    """Check same-node P2P health; raise RuntimeError on a detected TCP
    fallback unless MPIAMD_ALLOW_TCP=1. Returns a report dict."""
    report = {"checked": False, "groups": [], "busbw_gb": None, "problems": []}
    if not (dist.is_available() and dist.is_initialized()):
        return report
    if dist.get_world_size(group) <= 1 or not torch.cuda.is_available():
        return report
    report["checked"] = True
    groups = same_node_groups(group)
    report["groups"] = groups
    multi = [g for g in groups if len(g) > 1]
    if not multi:
        return report  # one rank per node: no same-node P2P to verify
    report["problems"] = static_preconditions()
    if not report["problems"]:
        # the canary runs on the WHOLE group (every rank must participate in
        # a collective); with any same-node pair present, a TCP fallback
        # caps the measured busbw far below one xGMI link
        bw = measure_busbw(group)
        report["busbw_gb"] = round(bw, 1)
        if bw < min_busbw:
            report["problems"].append(
                f"allreduce busbw {bw:.1f} GB/s < {min_busbw} GB/s floor for "
                "same-node ranks — RCCL is NOT using xGMI P2P (TCP/shm fallback)")
    if report["problems"]:
        msg = ("same-node P2P verification failed:\n  - " +
               "\n  - ".join(report["problems"]) +
               "\n(set MPIAMD_ALLOW_TCP=1 to run anyway)")
        if os.environ.get("MPIAMD_ALLOW_TCP") == "1":
            warnings.warn(msg)
        else:
            raise RuntimeError(msg)
    return report
