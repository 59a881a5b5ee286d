"""ValidateMPIJob — parity with reference
pkg/apis/kubeflow/validation/validation.go:49-160."""
from __future__ import annotations

import re

from . import constants as c
from . import types as t

_DNS1035 = re.compile(r"^[a-z]([-a-z0-9]*[a-z0-9])?$")
_VALID_RESTART = (c.RESTART_POLICY_NEVER, c.RESTART_POLICY_ON_FAILURE)
MULTIKUEUE_CONTROLLER = "kueue.x-k8s.io/multikueue"
_VALID_MANAGED_BY = (c.KUBEFLOW_JOB_CONTROLLER, MULTIKUEUE_CONTROLLER)


def _dns1035(label: str) -> bool:
    return len(label) <= 63 and bool(_DNS1035.match(label))


def validate_mpijob(job: dict) -> list[str]:
    """Returns a list of error strings (empty = valid)."""
    errs: list[str] = []
    errs += _validate_name(job)
    errs += _validate_spec(t.spec(job), "spec")
    return errs


def _validate_name(job: dict) -> list[str]:
    replicas = 1
    w = t.worker_spec(job)
    if w is not None and int(w.get("replicas") or 0) > 0:
        replicas = int(w["replicas"])
    hostname = f"{t.name(job)}-worker-{replicas - 1}"
    if not _dns1035(hostname):
        return [f"metadata.name: Invalid value: will not able to create pod "
                f"and service with invalid DNS label {hostname!r}"]
    return []


def _validate_spec(spec: dict, path: str) -> list[str]:
    errs = _validate_replica_specs(spec.get("mpiReplicaSpecs"), f"{path}.mpiReplicaSpecs")
    if spec.get("slotsPerWorker") is None:
        errs.append(f"{path}.slotsPerWorker: Required: must have number of slots per worker")
    elif int(spec["slotsPerWorker"]) < 0:
        errs.append(f"{path}.slotsPerWorker: must be greater than or equal to 0")
    errs += _validate_run_policy(spec.get("runPolicy", {}), f"{path}.runPolicy")
    if not spec.get("sshAuthMountPath"):
        errs.append(f"{path}.sshAuthMountPath: Required: must have a mount path for SSH credentials")
    if spec.get("mpiImplementation") not in c.MPI_IMPLEMENTATIONS:
        errs.append(f"{path}.mpiImplementation: Unsupported value: "
                    f"{spec.get('mpiImplementation')!r}: supported: {list(c.MPI_IMPLEMENTATIONS)}")
    return errs


def _validate_run_policy(rp: dict, path: str) -> list[str]:
    errs = []
    if rp.get("cleanPodPolicy") is None:
        errs.append(f"{path}.cleanPodPolicy: Required: must have clean Pod policy")
    elif rp["cleanPodPolicy"] not in c.CLEAN_POD_POLICIES:
        errs.append(f"{path}.cleanPodPolicy: Unsupported value: {rp['cleanPodPolicy']!r}")
    for field_name in ("ttlSecondsAfterFinished", "activeDeadlineSeconds", "backoffLimit"):
        v = rp.get(field_name)
        if v is not None and int(v) < 0:
            errs.append(f"{path}.{field_name}: must be greater than or equal to 0")
    mb = rp.get("managedBy")
    if mb is not None and mb not in _VALID_MANAGED_BY:
        errs.append(f"{path}.managedBy: Unsupported value: {mb!r}")
    return errs


def _validate_replica_specs(replicas, path: str) -> list[str]:
    if replicas is None:
        return [f"{path}: Required: must have replica specs"]
    errs = []
    launcher = replicas.get(c.MPI_REPLICA_TYPE_LAUNCHER)
    lpath = f"{path}[Launcher]"
    if launcher is None:
        errs.append(f"{lpath}: Required: must have Launcher replica spec")
    else:
        errs += _validate_replica(launcher, lpath)
        if launcher.get("replicas") is not None and int(launcher["replicas"]) != 1:
            errs.append(f"{lpath}.replicas: Invalid value: must be 1")
    worker = replicas.get(c.MPI_REPLICA_TYPE_WORKER)
    if worker is not None:
        wpath = f"{path}[Worker]"
        errs += _validate_replica(worker, wpath)
        if worker.get("replicas") is not None and int(worker["replicas"]) <= 0:
            errs.append(f"{wpath}.replicas: Invalid value: must be greater than or equal to 1")
    return errs


def _validate_replica(rs: dict, path: str) -> list[str]:
    errs = []
    if rs.get("replicas") is None:
        errs.append(f"{path}.replicas: Required: must define number of replicas")
    if rs.get("restartPolicy") not in _VALID_RESTART:
        errs.append(f"{path}.restartPolicy: Unsupported value: {rs.get('restartPolicy')!r}")
    if not t.deep_get(rs, "template", "spec", "containers", default=[]):
        errs.append(f"{path}.template.spec.containers: Required: must define at least one container")
    return errs
