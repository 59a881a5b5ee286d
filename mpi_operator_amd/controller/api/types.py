"""Typed accessors over MPIJob v2beta1 dicts (k8s JSON form).

The controller works on plain dicts (the natural Python representation of
kubernetes objects — the analog of the reference's generated Go structs,
reference pkg/apis/kubeflow/v2beta1/types.go). These helpers centralize the
spec fields the controller interprets."""
from __future__ import annotations

import datetime
from typing import Any

from . import constants as c

Obj = dict  # a k8s object in JSON form


def meta(o: Obj) -> dict:
    return o.setdefault("metadata", {})


def name(o: Obj) -> str:
    return meta(o).get("name", "")


def namespace(o: Obj) -> str:
    return meta(o).get("namespace", "default")


def uid(o: Obj) -> str:
    return meta(o).get("uid", "")


def spec(o: Obj) -> dict:
    return o.setdefault("spec", {})


def status(o: Obj) -> dict:
    return o.setdefault("status", {})


def run_policy(job: Obj) -> dict:
    return spec(job).setdefault("runPolicy", {})


def replica_specs(job: Obj) -> dict:
    return spec(job).setdefault("mpiReplicaSpecs", {})


def launcher_spec(job: Obj) -> dict | None:
    return replica_specs(job).get(c.MPI_REPLICA_TYPE_LAUNCHER)


def worker_spec(job: Obj) -> dict | None:
    return replica_specs(job).get(c.MPI_REPLICA_TYPE_WORKER)


def worker_replicas(job: Obj) -> int:
    w = worker_spec(job)
    if w is None:
        return 0
    return int(w.get("replicas", 0))


def slots_per_worker(job: Obj) -> int:
    return int(spec(job).get("slotsPerWorker", 1))


def run_launcher_as_worker(job: Obj) -> bool:
    return bool(spec(job).get("runLauncherAsWorker", False))


def mpi_implementation(job: Obj) -> str:
    return spec(job).get("mpiImplementation", c.MPI_IMPL_OPENMPI)


def ssh_auth_mount_path(job: Obj) -> str:
    return spec(job).get("sshAuthMountPath", c.ROOT_SSH_PATH)


def launcher_creation_policy(job: Obj) -> str:
    return spec(job).get("launcherCreationPolicy", c.LAUNCHER_CREATION_AT_STARTUP)


def is_suspended(job: Obj) -> bool:
    return bool(run_policy(job).get("suspend", False))


def managed_by(job: Obj) -> str:
    return run_policy(job).get("managedBy", c.KUBEFLOW_JOB_CONTROLLER)


def clean_pod_policy(job: Obj) -> str:
    return run_policy(job).get("cleanPodPolicy", c.CLEAN_POD_POLICY_NONE)


def scheduling_policy(job: Obj) -> dict | None:
    return run_policy(job).get("schedulingPolicy")


# ---------------- names ----------------
def launcher_name(job: Obj) -> str:
    return name(job) + c.LAUNCHER_SUFFIX


def worker_name(job: Obj, index: int) -> str:
    return f"{name(job)}{c.WORKER_SUFFIX}-{index}"


def config_name(job: Obj) -> str:
    return name(job) + c.CONFIG_SUFFIX


def ssh_secret_name(job: Obj) -> str:
    return name(job) + c.SSH_AUTH_SECRET_SUFFIX


# ---------------- conditions / status ----------------
def now_iso() -> str:
    return datetime.datetime.now(datetime.timezone.utc).strftime("%Y-%m-%dT%H:%M:%SZ")


def get_condition(st: dict, cond_type: str) -> dict | None:
    for cond in st.get("conditions", []):
        if cond.get("type") == cond_type:
            return cond
    return None


def has_condition_true(st: dict, cond_type: str) -> bool:
    cond = get_condition(st, cond_type)
    return cond is not None and cond.get("status") == "True"


def job_finished(job: Obj) -> bool:
    st = status(job)
    return has_condition_true(st, c.JOB_SUCCEEDED) or has_condition_true(st, c.JOB_FAILED)


def controlled_by(child: Obj, owner: Obj) -> bool:
    """metav1.IsControlledBy equivalent — controller ownerRef UID match."""
    for ref in meta(child).get("ownerReferences", []):
        if ref.get("controller") and ref.get("uid") == uid(owner):
            return True
    return False


def controller_ref(job: Obj) -> dict:
    return {
        "apiVersion": c.API_GROUP_VERSION,
        "kind": c.KIND,
        "name": name(job),
        "uid": uid(job),
        "controller": True,
        "blockOwnerDeletion": True,
    }


def deep_get(o: Any, *path, default=None):
    for p in path:
        if o is None:
            return default
        if isinstance(o, dict):
            o = o.get(p)
        else:
            try:
                o = o[p]
            except (IndexError, TypeError):
                return default
    return o if o is not None else default
