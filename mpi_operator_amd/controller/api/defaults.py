"""SetDefaults_MPIJob — parity with reference
pkg/apis/kubeflow/v2beta1/default.go:27-80."""
from __future__ import annotations

from . import constants as c
from . import types as t


def set_defaults_replica(rs: dict, default_restart: str, default_replicas: int) -> None:
    if rs.get("replicas") is None:
        rs["replicas"] = default_replicas
    if rs.get("restartPolicy") is None:
        rs["restartPolicy"] = default_restart


def set_defaults_mpijob(job: dict) -> None:
    spec = t.spec(job)
    if spec.get("slotsPerWorker") is None:
        spec["slotsPerWorker"] = 1
    rp = spec.setdefault("runPolicy", {})
    if rp.get("cleanPodPolicy") is None:
        rp["cleanPodPolicy"] = c.CLEAN_POD_POLICY_NONE
    if spec.get("sshAuthMountPath") is None:
        spec["sshAuthMountPath"] = c.ROOT_SSH_PATH
    if spec.get("mpiImplementation") is None:
        spec["mpiImplementation"] = c.MPI_IMPL_OPENMPI
    if spec.get("launcherCreationPolicy") is None:
        spec["launcherCreationPolicy"] = c.LAUNCHER_CREATION_AT_STARTUP
    replicas = spec.setdefault("mpiReplicaSpecs", {})
    launcher = replicas.get(c.MPI_REPLICA_TYPE_LAUNCHER)
    if launcher is not None:
        set_defaults_replica(launcher, c.RESTART_POLICY_ON_FAILURE, 1)
    worker = replicas.get(c.MPI_REPLICA_TYPE_WORKER)
    if worker is not None:
        set_defaults_replica(worker, c.RESTART_POLICY_NEVER, 0)
