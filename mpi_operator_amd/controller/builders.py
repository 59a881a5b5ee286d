"""Child-resource builders: ConfigMap (hostfile + discover_hosts.sh),
headless Service, SSH Secret, worker Pods, launcher Job.

Behavioral parity with reference pkg/controller/mpi_job_controller.go:
hostfile formats (:1335-1380), discover_hosts (:1383-1407), Service
(:1409-1438), Secret ECDSA P-521 (:1442-1477), worker Pod (:1499-1552),
launcher Job/pod template (:1554-1674), SSH volume (:1793-1816).

MI355X-native deltas (documented, deliberate): the launcher's GPU-hiding env
clears ROCR/HIP_VISIBLE_DEVICES instead of NVIDIA_* (reference :216-219,
1629-1635), and workers are expected to request `amd.com/gpu` resources.
"""
from __future__ import annotations

import copy
import subprocess
import tempfile
import os

from .api import constants as c
from .api import types as t


def default_labels(job_name: str, role: str) -> dict:
    return {
        c.OPERATOR_NAME_LABEL: c.OPERATOR_NAME,
        c.JOB_NAME_LABEL: job_name,
        c.JOB_ROLE_LABEL: role,
    }


def worker_selector(job_name: str) -> dict:
    return default_labels(job_name, c.WORKER)


def _domain_format(cluster_domain: str) -> str:
    fmt = "{}.{}.{}.svc"
    if cluster_domain:
        fmt += "." + cluster_domain
    return fmt


def new_config_map(job: dict, worker_replicas: int, cluster_domain: str = "") -> dict:
    """hostfile: OpenMPI `host slots=N`; Intel/MPICH `host:N` (reference
    :1347-1363) — byte-compatible."""
    slots = t.slots_per_worker(job)
    impl = t.mpi_implementation(job)
    dom = _domain_format(cluster_domain)
    lines = []

    def host_line(host):
        fqdn = dom.format(host, t.name(job), t.namespace(job))
        if impl == c.MPI_IMPL_OPENMPI:
            return f"{fqdn} slots={slots}\n"
        return f"{fqdn}:{slots}\n"

    if t.run_launcher_as_worker(job):
        lines.append(host_line(t.launcher_name(job)))
    for i in range(worker_replicas):
        lines.append(host_line(t.worker_name(job, i)))
    return {
        "apiVersion": "v1",
        "kind": "ConfigMap",
        "metadata": {
            "name": t.config_name(job),
            "namespace": t.namespace(job),
            "labels": {"app": t.name(job)},
            "ownerReferences": [t.controller_ref(job)],
        },
        "data": {c.HOSTFILE_NAME: "".join(lines)},
    }


def update_discover_hosts(config_map: dict, job: dict, running_worker_pods: list,
                          cluster_domain: str = "") -> None:
    """discover_hosts.sh from the sorted running worker pods (reference
    :1383-1407; elastic protocol proposals/elastic-horovod.md:19-31)."""
    dom = _domain_format(cluster_domain)
    lines = ["#!/bin/sh\n"]
    if t.run_launcher_as_worker(job):
        lines.append("echo {}\n".format(
            dom.format(t.launcher_name(job), t.name(job), t.namespace(job))))
    for p in sorted(running_worker_pods, key=t.name):
        lines.append("echo {}\n".format(
            dom.format(t.name(p), t.name(job), t.namespace(p))))
    config_map.setdefault("data", {})[c.DISCOVER_HOSTS_SCRIPT_NAME] = "".join(lines)


def new_job_service(job: dict) -> dict:
    selector = {
        c.OPERATOR_NAME_LABEL: c.OPERATOR_NAME,
        c.JOB_NAME_LABEL: t.name(job),
    }
    return {
        "apiVersion": "v1",
        "kind": "Service",
        "metadata": {
            "name": t.name(job),
            "namespace": t.namespace(job),
            "labels": {"app": t.name(job)},
            "ownerReferences": [t.controller_ref(job)],
        },
        "spec": {
            "clusterIP": "None",
            "selector": selector,
            # true only with runLauncherAsWorker to avoid launcher-ready
            # deadlock (reference :1430-1434)
            "publishNotReadyAddresses": t.run_launcher_as_worker(job),
        },
    }


def generate_ssh_keypair() -> tuple[bytes, bytes]:
    """ECDSA P-521 keypair: (private PEM, public in authorized_keys form) —
    same algorithm/format as the reference (:1442-1460), via ssh-keygen."""
    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "key")
        subprocess.run(["ssh-keygen", "-q", "-t", "ecdsa", "-b", "521", "-N", "",
                        "-m", "PEM", "-f", path], check=True,
                       stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
        with open(path, "rb") as f:
            priv = f.read()
        with open(path + ".pub", "rb") as f:
            pub = f.read()
    return priv, pub


def new_ssh_auth_secret(job: dict, keygen=generate_ssh_keypair) -> dict:
    import base64
    priv, pub = keygen()
    return {
        "apiVersion": "v1",
        "kind": "Secret",
        "metadata": {
            "name": t.ssh_secret_name(job),
            "namespace": t.namespace(job),
            "labels": {"app": t.name(job)},
            "ownerReferences": [t.controller_ref(job)],
        },
        "type": "kubernetes.io/ssh-auth",
        "data": {
            "ssh-privatekey": base64.b64encode(priv).decode(),
            c.SSH_PUBLIC_KEY: base64.b64encode(pub).decode(),
        },
    }


_SSH_VOLUME_ITEMS = [
    {"key": "ssh-privatekey", "path": c.SSH_PRIVATE_KEY_FILE},
    {"key": c.SSH_PUBLIC_KEY, "path": c.SSH_PUBLIC_KEY_FILE},
    {"key": c.SSH_PUBLIC_KEY, "path": c.SSH_AUTHORIZED_KEYS_FILE},
]

_CONFIG_VOLUME_ITEMS = [
    {"key": c.HOSTFILE_NAME, "path": c.HOSTFILE_NAME, "mode": 0o444},
    {"key": c.DISCOVER_HOSTS_SCRIPT_NAME, "path": c.DISCOVER_HOSTS_SCRIPT_NAME,
     "mode": 0o555},
]


def _setup_ssh_on_pod(pod_spec: dict, job: dict) -> None:
    mode = 0o600 if t.ssh_auth_mount_path(job) == c.ROOT_SSH_PATH else None
    vol = {
        "name": c.SSH_AUTH_VOLUME,
        "secret": {
            "secretName": t.ssh_secret_name(job),
            "items": copy.deepcopy(_SSH_VOLUME_ITEMS),
        },
    }
    if mode is not None:
        vol["secret"]["defaultMode"] = mode
    pod_spec.setdefault("volumes", []).append(vol)
    main = pod_spec["containers"][0]
    main.setdefault("volumeMounts", []).append({
        "name": c.SSH_AUTH_VOLUME,
        "mountPath": t.ssh_auth_mount_path(job),
    })


def _set_restart_policy(pod_spec: dict, replica_spec: dict) -> None:
    rp = replica_spec.get("restartPolicy", "")
    # ExitCode maps to Never at pod level (reference :1722-1728)
    pod_spec["restartPolicy"] = c.RESTART_POLICY_NEVER if rp == c.RESTART_POLICY_EXIT_CODE else rp


def worker_replica_index_label(job: dict, index: int) -> str:
    # pad by one with runLauncherAsWorker (Kueue TAS needs unique indices,
    # reference :1487-1494)
    return str(index + 1) if t.run_launcher_as_worker(job) else str(index)


def new_worker(job: dict, index: int, podgroup_ctrl=None) -> dict:
    name = t.worker_name(job, index)
    tmpl = copy.deepcopy(t.deep_get(t.worker_spec(job), "template", default={}))
    meta = tmpl.setdefault("metadata", {})
    labels = meta.setdefault("labels", {})
    labels.update(default_labels(t.name(job), c.WORKER))
    labels[c.REPLICA_INDEX_LABEL] = worker_replica_index_label(job, index)
    spec = tmpl.setdefault("spec", {})
    spec["hostname"] = name
    spec["subdomain"] = t.name(job)  # matches the job Service name
    if spec.get("hostNetwork"):
        spec["dnsPolicy"] = "ClusterFirstWithHostNet"
    search = f"{t.name(job)}.{t.namespace(job)}.svc.cluster.local"
    dns = spec.setdefault("dnsConfig", {})
    dns.setdefault("searches", []).append(search)
    _set_restart_policy(spec, t.worker_spec(job))
    container = spec["containers"][0]
    if not container.get("command") and not container.get("args"):
        container["command"] = ["/usr/sbin/sshd", "-De"]
    container.setdefault("env", []).append({"name": "K_MPI_JOB_ROLE", "value": c.WORKER})
    _setup_ssh_on_pod(spec, job)
    if podgroup_ctrl is not None:
        podgroup_ctrl.decorate_pod_template(tmpl, t.name(job))
    return {
        "apiVersion": "v1",
        "kind": "Pod",
        "metadata": {
            "name": name,
            "namespace": t.namespace(job),
            "labels": labels,
            "annotations": meta.get("annotations", {}),
            "ownerReferences": [t.controller_ref(job)],
        },
        "spec": spec,
    }


def new_launcher_pod_template(job: dict, podgroup_ctrl=None, warn_event=None) -> dict:
    launcher_name = t.launcher_name(job)
    tmpl = copy.deepcopy(t.deep_get(t.launcher_spec(job), "template", default={}))
    meta = tmpl.setdefault("metadata", {})
    labels = meta.setdefault("labels", {})
    labels.update(default_labels(t.name(job), c.LAUNCHER))
    spec = tmpl.setdefault("spec", {})
    if podgroup_ctrl is not None:
        podgroup_ctrl.decorate_pod_template(tmpl, t.name(job))
    if t.run_launcher_as_worker(job):
        labels[c.REPLICA_INDEX_LABEL] = "0"
    spec["hostname"] = launcher_name
    spec["subdomain"] = t.name(job)
    if spec.get("hostNetwork"):
        spec["dnsPolicy"] = "ClusterFirstWithHostNet"
    container = spec["containers"][0]
    env = container.setdefault("env", [])
    env.append({"name": "K_MPI_JOB_ROLE", "value": c.LAUNCHER})
    slots = str(t.slots_per_worker(job))
    impl = t.mpi_implementation(job)
    cfg = c.CONFIG_MOUNT_PATH
    if impl == c.MPI_IMPL_OPENMPI:
        env += [
            {"name": "OMPI_MCA_orte_keep_fqdn_hostnames", "value": "true"},
            {"name": "OMPI_MCA_orte_default_hostfile", "value": f"{cfg}/{c.HOSTFILE_NAME}"},
            {"name": "OMPI_MCA_plm_rsh_args", "value": "-o ConnectionAttempts=10"},
            {"name": c.OPENMPI_SLOTS_ENV, "value": slots},
        ]
    elif impl == c.MPI_IMPL_INTEL:
        env += [
            {"name": "I_MPI_HYDRA_HOST_FILE", "value": f"{cfg}/{c.HOSTFILE_NAME}"},
            {"name": "I_MPI_HYDRA_BOOTSTRAP_EXEC_EXTRA_ARGS",
             "value": "-o ConnectionAttempts=10"},
            {"name": c.INTELMPI_SLOTS_ENV, "value": slots},
        ]
    elif impl == c.MPI_IMPL_MPICH:
        env += [
            {"name": "HYDRA_HOST_FILE", "value": f"{cfg}/{c.HOSTFILE_NAME}"},
            {"name": "HYDRA_LAUNCH_EXTRA_ARGS", "value": "-o ConnectionAttempts=10"},
        ]
    if not t.run_launcher_as_worker(job):
        # hide the node's GPUs from a non-worker launcher — ROCm env
        # (MI355X-native replacement for the reference's NVIDIA_* clearing)
        env += [{"name": n, "value": ""} for n in c.AMD_DISABLE_GPU_ENV]
    _setup_ssh_on_pod(spec, job)
    if spec.get("restartPolicy"):
        if warn_event is not None:
            warn_event("SetPodTemplateRestartPolicy",
                       "Restart policy in pod template overridden by restart policy in replica spec")
    _set_restart_policy(spec, t.launcher_spec(job))
    spec.setdefault("volumes", []).append({
        "name": c.CONFIG_VOLUME_NAME,
        "configMap": {
            "name": t.config_name(job),
            "items": copy.deepcopy(_CONFIG_VOLUME_ITEMS),
        },
    })
    container.setdefault("volumeMounts", []).append({
        "name": c.CONFIG_VOLUME_NAME,
        "mountPath": cfg,
    })
    return {
        "metadata": {
            "labels": labels,
            "annotations": meta.get("annotations", {}),
            "ownerReferences": [t.controller_ref(job)],
        },
        "spec": spec,
    }


def new_launcher_job(job: dict, podgroup_ctrl=None, warn_event=None) -> dict:
    rp = t.run_policy(job)
    job_spec = {
        "template": new_launcher_pod_template(job, podgroup_ctrl, warn_event),
        "podReplacementPolicy": "Failed",
    }
    for src, dst in (("ttlSecondsAfterFinished", "ttlSecondsAfterFinished"),
                     ("activeDeadlineSeconds", "activeDeadlineSeconds"),
                     ("backoffLimit", "backoffLimit")):
        if rp.get(src) is not None:
            job_spec[dst] = rp[src]
    if t.is_suspended(job):
        job_spec["suspend"] = True
    return {
        "apiVersion": "batch/v1",
        "kind": "Job",
        "metadata": {
            "name": t.launcher_name(job),
            "namespace": t.namespace(job),
            "labels": {"app": t.name(job)},
            "ownerReferences": [t.controller_ref(job)],
        },
        "spec": job_spec,
    }
