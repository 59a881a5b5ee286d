"""Controller unit tests against the fake client — mirrors the reference's
fake-clientset test matrix (reference pkg/controller/mpi_job_controller_test.go,
SURVEY.md §4.1)."""
import base64
import copy

import pytest

from mpi_operator_amd.controller import MPIJobController, VolcanoCtrl, SchedulerPluginsCtrl
from mpi_operator_amd.controller.api import constants as c
from mpi_operator_amd.controller.api import defaults, types as t, validation
from mpi_operator_amd.controller import builders
from mpi_operator_amd.controller.client import FakeKubeClient
from mpi_operator_amd.controller.client.base import JOBS, MPIJOBS, PODS, PODGROUPS_VOLCANO


def fake_keygen():
    return b"PRIVATE", b"ecdsa-sha2-nistp521 AAAA test"


def make_job(name="test", namespace="default", workers=2, **spec_over):
    job = {
        "apiVersion": c.API_GROUP_VERSION,
        "kind": "MPIJob",
        "metadata": {"name": name, "namespace": namespace},
        "spec": {
            "mpiReplicaSpecs": {
                "Launcher": {
                    "template": {"spec": {"containers": [
                        {"name": "launcher", "image": "mpi-amd:latest",
                         "command": ["mpirun", "python", "train.py"]}]}},
                },
                "Worker": {
                    "replicas": workers,
                    "template": {"spec": {"containers": [
                        {"name": "worker", "image": "mpi-amd:latest",
                         "resources": {"limits": {"amd.com/gpu": 8}}}]}},
                },
            },
        },
    }
    job["spec"].update(spec_over)
    return job


def make_controller(podgroup=None, **kw):
    client = FakeKubeClient()
    ctrl = MPIJobController(client, podgroup_ctrl=podgroup,
                            keygen=fake_keygen, **kw)
    return client, ctrl


def seed_and_sync(client, ctrl, job):
    stored = client.seed(MPIJOBS, job)
    client.clear_actions()
    ctrl.sync(t.namespace(stored), t.name(stored))
    return stored


# ------------------------- resource creation -------------------------
def test_all_resources_created():
    client, ctrl = make_controller()
    job = seed_and_sync(client, ctrl, make_job(workers=2))
    ns = "default"
    svc = client.services.get(ns, "test")
    assert svc["spec"]["clusterIP"] == "None"
    assert svc["spec"]["publishNotReadyAddresses"] is False
    cm = client.configmaps.get(ns, "test-config")
    assert cm["data"]["hostfile"] == (
        "test-worker-0.test.default.svc slots=1\n"
        "test-worker-1.test.default.svc slots=1\n")
    assert cm["data"]["discover_hosts.sh"] == "#!/bin/sh\n"
    secret = client.secrets.get(ns, "test-ssh")
    assert secret["type"] == "kubernetes.io/ssh-auth"
    assert base64.b64decode(secret["data"]["ssh-privatekey"]) == b"PRIVATE"
    for i in range(2):
        pod = client.pods.get(ns, f"test-worker-{i}")
        assert pod["spec"]["hostname"] == f"test-worker-{i}"
        assert pod["spec"]["subdomain"] == "test"
        assert pod["spec"]["containers"][0]["command"] == ["/usr/sbin/sshd", "-De"]
        assert pod["metadata"]["labels"][c.REPLICA_INDEX_LABEL] == str(i)
    launcher = client.jobs.get(ns, "test-launcher")
    assert launcher["spec"]["podReplacementPolicy"] == "Failed"
    env = {e["name"]: e.get("value") for e in
           launcher["spec"]["template"]["spec"]["containers"][0]["env"]}
    assert env["OMPI_MCA_orte_default_hostfile"] == "/etc/mpi/hostfile"
    assert env[c.OPENMPI_SLOTS_ENV] == "1"
    assert env["ROCR_VISIBLE_DEVICES"] == ""  # launcher GPUs hidden
    # status updated with Created condition
    updated = client.mpijobs.get(ns, "test")
    assert t.get_condition(updated["status"], c.JOB_CREATED)["status"] == "True"


def test_hostfile_intel_format_and_launcher_as_worker():
    client, ctrl = make_controller()
    job = make_job(workers=1, mpiImplementation="Intel", runLauncherAsWorker=True,
                   slotsPerWorker=4)
    seed_and_sync(client, ctrl, job)
    cm = client.configmaps.get("default", "test-config")
    assert cm["data"]["hostfile"] == (
        "test-launcher.test.default.svc:4\n"
        "test-worker-0.test.default.svc:4\n")
    svc = client.services.get("default", "test")
    assert svc["spec"]["publishNotReadyAddresses"] is True
    # worker index labels padded by one
    pod = client.pods.get("default", "test-worker-0")
    assert pod["metadata"]["labels"][c.REPLICA_INDEX_LABEL] == "1"


def test_cluster_domain_in_hostfile():
    client, ctrl = make_controller(cluster_domain="cluster.local")
    seed_and_sync(client, ctrl, make_job(workers=1))
    cm = client.configmaps.get("default", "test-config")
    assert "test-worker-0.test.default.svc.cluster.local slots=1" in cm["data"]["hostfile"]


def test_discover_hosts_running_workers_only():
    client, ctrl = make_controller()
    job = client.seed(MPIJOBS, make_job(workers=2))
    ctrl.sync("default", "test")
    # mark worker-1 running (out of order to test sorting), worker-0 pending
    p1 = client.pods.get("default", "test-worker-1")
    p1["status"] = {"phase": "Running"}
    client.pods.update("default", p1)
    p0 = client.pods.get("default", "test-worker-0")
    p0["status"] = {"phase": "Pending"}
    client.pods.update("default", p0)
    ctrl.sync("default", "test")
    cm = client.configmaps.get("default", "test-config")
    assert cm["data"]["discover_hosts.sh"] == \
        "#!/bin/sh\necho test-worker-1.test.default.svc\n"


# ------------------------- ownership refusal -------------------------
@pytest.mark.parametrize("kind,seed_fn", [
    ("Service", lambda cl: cl.services.create("default", {
        "metadata": {"name": "test", "namespace": "default"}, "spec": {}})),
    ("ConfigMap", lambda cl: cl.configmaps.create("default", {
        "metadata": {"name": "test-config", "namespace": "default"}, "data": {}})),
    ("Secret", lambda cl: cl.secrets.create("default", {
        "metadata": {"name": "test-ssh", "namespace": "default"}, "data": {}})),
    ("Job", lambda cl: cl.jobs.create("default", {
        "metadata": {"name": "test-launcher", "namespace": "default"}, "spec": {}})),
])
def test_ownership_conflict_refused(kind, seed_fn):
    client, ctrl = make_controller()
    seed_fn(client)  # pre-existing object NOT owned by the job
    client.seed(MPIJOBS, make_job())
    with pytest.raises(RuntimeError, match="not managed by MPIJob"):
        ctrl.sync("default", "test")
    evs = client.events.list("default")
    assert any(e["reason"] == c.ERR_RESOURCE_EXISTS for e in evs)


# ------------------------- scale-down (elastic) -------------------------
def test_worker_scale_down_deletes_high_indices():
    client, ctrl = make_controller()
    job = client.seed(MPIJOBS, make_job(workers=3))
    ctrl.sync("default", "test")
    assert len(client.pods.list("default", builders.worker_selector("test"))) == 3
    job = client.mpijobs.get("default", "test")
    job["spec"]["mpiReplicaSpecs"]["Worker"]["replicas"] = 1
    client.mpijobs.update("default", job)
    ctrl.sync("default", "test")
    pods = client.pods.list("default", builders.worker_selector("test"))
    assert [t.name(p) for p in pods] == ["test-worker-0"]


# ------------------------- suspend / resume -------------------------
def test_suspend_creates_suspended_launcher_no_workers():
    client, ctrl = make_controller()
    job = make_job(workers=2)
    job["spec"]["runPolicy"] = {"suspend": True}
    seed_and_sync(client, ctrl, job)
    launcher = client.jobs.get("default", "test-launcher")
    assert launcher["spec"]["suspend"] is True
    assert client.pods.list("default", builders.worker_selector("test")) == []
    updated = client.mpijobs.get("default", "test")
    assert t.get_condition(updated["status"], c.JOB_SUSPENDED)["status"] == "True"
    assert updated["status"].get("startTime") is None


def test_resume_clears_job_start_time_and_unsuspends():
    client, ctrl = make_controller()
    job = make_job(workers=1)
    job["spec"]["runPolicy"] = {"suspend": True}
    client.seed(MPIJOBS, job)
    ctrl.sync("default", "test")
    # simulate kubelet having set the launcher startTime while suspended once
    launcher = client.jobs.get("default", "test-launcher")
    launcher["status"] = {"startTime": "2020-01-01T00:00:00Z"}
    client.jobs.update("default", launcher)
    # resume
    job = client.mpijobs.get("default", "test")
    job["spec"]["runPolicy"]["suspend"] = False
    client.mpijobs.update("default", job)
    ctrl.sync("default", "test")
    launcher = client.jobs.get("default", "test-launcher")
    assert launcher["spec"]["suspend"] is False
    assert launcher["status"].get("startTime") is None  # cleared via subresource
    updated = client.mpijobs.get("default", "test")
    cond = t.get_condition(updated["status"], c.JOB_SUSPENDED)
    assert cond["status"] == "False" and cond["reason"] == "MPIJobResumed"
    assert updated["status"].get("startTime") is not None


def test_suspend_running_job_deletes_workers():
    client, ctrl = make_controller()
    client.seed(MPIJOBS, make_job(workers=2))
    ctrl.sync("default", "test")
    assert len(client.pods.list("default", builders.worker_selector("test"))) == 2
    job = client.mpijobs.get("default", "test")
    job["spec"]["runPolicy"] = {"suspend": True}
    client.mpijobs.update("default", job)
    ctrl.sync("default", "test")
    assert client.pods.list("default", builders.worker_selector("test")) == []
    launcher = client.jobs.get("default", "test-launcher")
    assert launcher["spec"]["suspend"] is True


# ------------------------- WaitForWorkersReady -------------------------
def test_wait_for_workers_ready_gates_launcher():
    client, ctrl = make_controller()
    job = make_job(workers=2, launcherCreationPolicy="WaitForWorkersReady")
    client.seed(MPIJOBS, job)
    ctrl.sync("default", "test")
    from mpi_operator_amd.controller.client.base import NotFound
    with pytest.raises(NotFound):
        client.jobs.get("default", "test-launcher")
    # make both workers Ready
    for i in range(2):
        p = client.pods.get("default", f"test-worker-{i}")
        p["status"] = {"phase": "Running",
                       "conditions": [{"type": "Ready", "status": "True"}]}
        client.pods.update("default", p)
    ctrl.sync("default", "test")
    assert client.jobs.get("default", "test-launcher") is not None


# ------------------------- completion / failure -------------------------
def _complete_launcher(client, succeeded=True):
    launcher = client.jobs.get("default", "test-launcher")
    cond = {"type": "Complete" if succeeded else "Failed", "status": "True"}
    if not succeeded:
        cond["reason"] = "BackoffLimitExceeded"
    launcher["status"] = {"conditions": [cond],
                          "completionTime": "2026-01-01T00:00:00Z"}
    if not succeeded:
        launcher["status"]["failed"] = 1
    client.jobs.update("default", launcher)


def test_launcher_succeeded_marks_job_succeeded():
    client, ctrl = make_controller()
    client.seed(MPIJOBS, make_job(workers=1))
    ctrl.sync("default", "test")
    _complete_launcher(client, succeeded=True)
    ctrl.sync("default", "test")
    job = client.mpijobs.get("default", "test")
    assert t.has_condition_true(job["status"], c.JOB_SUCCEEDED)
    assert job["status"]["completionTime"] == "2026-01-01T00:00:00Z"
    assert job["status"]["replicaStatuses"]["Launcher"]["succeeded"] == 1


def test_launcher_failed_backoff_concatenates_last_pod():
    client, ctrl = make_controller()
    client.seed(MPIJOBS, make_job(workers=1))
    ctrl.sync("default", "test")
    launcher = client.jobs.get("default", "test-launcher")
    launcher["spec"]["selector"] = {"matchLabels": {"controller-uid": "xyz"}}
    client.jobs.update("default", launcher)
    client.pods.create("default", {
        "metadata": {"name": "test-launcher-abc", "namespace": "default",
                     "labels": {"controller-uid": "xyz"},
                     "creationTimestamp": "2026-01-01T00:00:01Z"},
        "status": {"phase": "Failed", "reason": "OOMKilled",
                   "message": "launcher OOM"}})
    _complete_launcher(client, succeeded=False)
    ctrl.sync("default", "test")
    job = client.mpijobs.get("default", "test")
    cond = t.get_condition(job["status"], c.JOB_FAILED)
    assert cond["status"] == "True"
    assert cond["reason"] == "BackoffLimitExceeded/OOMKilled"
    assert "launcher OOM" in cond["message"]


def test_clean_pod_policy_running_after_finish():
    client, ctrl = make_controller()
    job = make_job(workers=2)
    job["spec"]["runPolicy"] = {"cleanPodPolicy": "Running"}
    client.seed(MPIJOBS, job)
    ctrl.sync("default", "test")
    # worker-0 running (deleted), worker-1 succeeded (kept under Running policy)
    p = client.pods.get("default", "test-worker-0")
    p["status"] = {"phase": "Running"}
    client.pods.update("default", p)
    p = client.pods.get("default", "test-worker-1")
    p["status"] = {"phase": "Succeeded"}
    client.pods.update("default", p)
    _complete_launcher(client)
    ctrl.sync("default", "test")  # marks finished
    ctrl.sync("default", "test")  # cleanup pass
    names = [t.name(p) for p in client.pods.list("default", builders.worker_selector("test"))]
    assert names == ["test-worker-1"]


# ------------------------- managedBy -------------------------
def test_managed_by_external_controller_skipped():
    client, ctrl = make_controller()
    job = make_job()
    job["spec"]["runPolicy"] = {"managedBy": "kueue.x-k8s.io/multikueue"}
    client.seed(MPIJOBS, job)
    client.clear_actions()
    ctrl.sync("default", "test")
    assert client.actions_of("create") == []  # nothing reconciled


# ------------------------- validation events -------------------------
def test_invalid_job_emits_warning_no_requeue():
    client, ctrl = make_controller()
    job = make_job()
    del job["spec"]["mpiReplicaSpecs"]["Launcher"]
    client.seed(MPIJOBS, job)
    ctrl.sync("default", "test")
    evs = client.events.list("default")
    assert any(e["reason"] == c.VALIDATION_ERROR for e in evs)
    assert client.actions_of("create", "pods") == []


def test_wait_for_workers_ready_suspended_does_not_create_launcher():
    """Suspended + WaitForWorkersReady: no workers exist, so the launcher
    must NOT be created (the gate counts desired replicas, not the empty
    created list)."""
    from mpi_operator_amd.controller.client.base import NotFound
    client, ctrl = make_controller()
    seed_and_sync(client, ctrl, make_job(
        launcherCreationPolicy="WaitForWorkersReady",
        runPolicy={"suspend": True}))
    with pytest.raises(NotFound):
        client.jobs.get("default", "test-launcher")
