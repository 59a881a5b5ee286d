"""Integration tier — the reference's envtest equivalent:
a LIVE OperatorServer (watch loops + worker threads) reconciles MPIJobs
against the fake apiserver, and — because there is no kubelet — the test
plays kubelet: it patches worker pods to Running, fabricates a launcher pod
for the batch Job, and completes the Job by setting its conditions
(reference test/integration/mpi_job_controller_test.go:153-166,1165,1281)."""
import threading
import time

import pytest

from mpi_operator_amd.controller.api import constants as c
from mpi_operator_amd.controller.client.base import JOBS, MPIJOBS, PODS
from mpi_operator_amd.controller.client.fake import FakeKubeClient
from mpi_operator_amd.controller.reconciler import MPIJobController
from mpi_operator_amd.controller.server import OperatorServer

NS = "it"


def mk_job(name, workers=2, **spec_extra):
    spec = {
        "slotsPerWorker": 1,
        "mpiReplicaSpecs": {
            "Launcher": {"replicas": 1, "template": {"spec": {"containers": [
                {"name": "l", "image": "img", "command": ["amdrun"]}]}}},
            "Worker": {"replicas": workers, "template": {"spec": {"containers": [
                {"name": "w", "image": "img"}]}}},
        },
    }
    spec.update(spec_extra)
    return {"apiVersion": c.API_GROUP_VERSION, "kind": c.KIND,
            "metadata": {"name": name, "namespace": NS}, "spec": spec}


@pytest.fixture()
def live():
    client = FakeKubeClient()
    controller = MPIJobController(client)
    server = OperatorServer(client, controller, namespace=NS, threadiness=2, resync_s=1)
    server.run()
    yield client, server
    server.stop.set()
    time.sleep(0.1)


def wait_for(predicate, timeout=10.0, what="condition"):
    deadline = time.time() + timeout
    while time.time() < deadline:
        v = predicate()
        if v:
            return v
        time.sleep(0.05)
    raise AssertionError(f"timed out waiting for {what}")


def get_job(client, name):
    return client.mpijobs.get(NS, name)


def has_cond(job, cond_type, status="True"):
    for cond in job.get("status", {}).get("conditions", []):
        if cond["type"] == cond_type and cond["status"] == status:
            return True
    return False


def worker_pods(client, name):
    return [p for p in client.pods.list(NS)
            if p["metadata"].get("labels", {}).get(c.JOB_ROLE_LABEL) == c.WORKER
            and p["metadata"].get("labels", {}).get(c.JOB_NAME_LABEL) == name]


def kubelet_run_workers(client, name):
    """Mock kubelet: mark every worker pod Running+Ready."""
    for p in worker_pods(client, name):
        p["status"] = {"phase": "Running",
                       "conditions": [{"type": "Ready", "status": "True"}]}
        client.pods.update(NS, p)


def kubelet_start_launcher(client, name):
    """Mock kubelet: fabricate a Running launcher pod for the batch Job
    (reference createPodForJob, test/integration/...:1281)."""
    job = wait_for(lambda: _try(lambda: client.jobs.get(NS, name + "-launcher")),
                   what="launcher Job")
    pod = {"metadata": {"name": name + "-launcher-xyz", "namespace": NS,
                        "labels": {c.JOB_NAME_LABEL: name},
                        "ownerReferences": [{"kind": "Job", "name": name + "-launcher",
                                             "uid": job["metadata"]["uid"],
                                             "controller": True}]},
           "spec": {}, "status": {"phase": "Running"}}
    try:
        client.pods.create(NS, pod)
    except Exception:
        pass
    return pod


def kubelet_complete_launcher(client, name, succeed=True):
    """Mock kubelet: finish the launcher pod and complete the batch Job
    (reference sets JobSuccessCriteriaMet+JobComplete, ...:153-166)."""
    job = wait_for(lambda: _try(lambda: client.jobs.get(NS, name + "-launcher")),
                   what="launcher Job")
    pod = _try(lambda: client.pods.get(NS, name + "-launcher-xyz"))
    if pod is not None:
        pod["status"]["phase"] = "Succeeded" if succeed else "Failed"
        client.pods.update(NS, pod)
    cond = {"type": "Complete" if succeed else "Failed", "status": "True",
            "reason": "" if succeed else "BackoffLimitExceeded"}
    job.setdefault("status", {})["conditions"] = [cond]
    if succeed:
        job["status"]["succeeded"] = 1
    else:
        job["status"]["failed"] = 1
    client.jobs.update_status(NS, job)


def _try(fn):
    try:
        return fn()
    except Exception:
        return None


def test_job_runs_to_success(live):
    client, _ = live
    client.mpijobs.create(NS, mk_job("ok"))
    # controller must create svc/cm/secret/workers/launcher on its own
    wait_for(lambda: len(worker_pods(client, "ok")) == 2, what="worker pods")
    wait_for(lambda: _try(lambda: client.configmaps.get(NS, "ok-config")), what="configmap")
    wait_for(lambda: _try(lambda: client.secrets.get(NS, "ok-ssh")), what="ssh secret")
    kubelet_run_workers(client, "ok")
    kubelet_start_launcher(client, "ok")
    wait_for(lambda: has_cond(get_job(client, "ok"), c.JOB_RUNNING), what="Running")
    kubelet_complete_launcher(client, "ok", succeed=True)
    wait_for(lambda: has_cond(get_job(client, "ok"), c.JOB_SUCCEEDED), what="Succeeded")


def test_job_failure_propagates(live):
    client, _ = live
    client.mpijobs.create(NS, mk_job("bad"))
    wait_for(lambda: len(worker_pods(client, "bad")) == 2, what="worker pods")
    kubelet_run_workers(client, "bad")
    kubelet_complete_launcher(client, "bad", succeed=False)
    wait_for(lambda: has_cond(get_job(client, "bad"), c.JOB_FAILED), what="Failed")


def test_wait_for_workers_ready_gates_launcher(live):
    client, _ = live
    client.mpijobs.create(NS, mk_job("gated", launcherCreationPolicy="WaitForWorkersReady"))
    wait_for(lambda: len(worker_pods(client, "gated")) == 2, what="worker pods")
    time.sleep(0.5)  # give the controller a chance to (wrongly) create it
    assert _try(lambda: client.jobs.get(NS, "gated-launcher")) is None
    kubelet_run_workers(client, "gated")
    wait_for(lambda: _try(lambda: client.jobs.get(NS, "gated-launcher")),
             what="launcher after workers Ready")


def test_suspend_deletes_workers_resume_recreates(live):
    client, _ = live
    client.mpijobs.create(NS, mk_job("s"))
    wait_for(lambda: len(worker_pods(client, "s")) == 2, what="worker pods")
    job = get_job(client, "s")
    job["spec"]["runPolicy"] = {"suspend": True}
    client.mpijobs.update(NS, job)
    wait_for(lambda: len(worker_pods(client, "s")) == 0, what="workers gone")
    wait_for(lambda: has_cond(get_job(client, "s"), c.JOB_SUSPENDED), what="Suspended")
    job = get_job(client, "s")
    job["spec"]["runPolicy"] = {"suspend": False}
    client.mpijobs.update(NS, job)
    wait_for(lambda: len(worker_pods(client, "s")) == 2, what="workers back")
    wait_for(lambda: has_cond(get_job(client, "s"), c.JOB_SUSPENDED, "False"),
             what="unsuspended")


def test_worker_scale_down_elastic(live):
    client, _ = live
    client.mpijobs.create(NS, mk_job("el", workers=4))
    wait_for(lambda: len(worker_pods(client, "el")) == 4, what="4 workers")
    job = get_job(client, "el")
    job["spec"]["mpiReplicaSpecs"]["Worker"]["replicas"] = 2
    client.mpijobs.update(NS, job)
    wait_for(lambda: len(worker_pods(client, "el")) == 2, what="scaled to 2")
    names = sorted(p["metadata"]["name"] for p in worker_pods(client, "el"))
    assert names == ["el-worker-0", "el-worker-1"]
