"""Ports of the reference's deep controller test matrix:

- suspended-create per MPI implementation
  (reference mpi_job_controller_test.go:961 TestCreateSuspendedMPIJob)
- suspend-while-running semantics (:1010 TestSuspendedRunningMPIJob)
- resume with Kueue-injected scheduling directives, KEP-2926
  (:1207 TestResumeMPIJobWithExistingLauncher)
- failed launcher Update must not poison the controller's view
  (:1163 TestUnsuspendLauncherUpdateFailureDoesNotPoisonCache)
- WaitForWorkersReady / replica-status matrix
  (:1415 TestLauncherActiveWorkerNotReady, :1468 TestLauncherActiveWorkerReady)
- podgroup minResources cases (podgroup_test.go:442 TestCalculatePGMinResources)
"""
import pytest

from mpi_operator_amd.controller import MPIJobController
from mpi_operator_amd.controller.api import constants as c
from mpi_operator_amd.controller.api import types as t
from mpi_operator_amd.controller import builders
from mpi_operator_amd.controller.client import FakeKubeClient
from mpi_operator_amd.controller.client.base import MPIJOBS, NotFound, ApiError

from tests.test_controller import fake_keygen, make_job


def make_controller(**kw):
    client = FakeKubeClient()
    ctrl = MPIJobController(client, keygen=fake_keygen, **kw)
    return client, ctrl


# ---- TestCreateSuspendedMPIJob (:961), per implementation ----
@pytest.mark.parametrize("impl", ["OpenMPI", "Intel", "MPICH"])
def test_create_suspended_mpijob_per_impl(impl):
    client, ctrl = make_controller()
    job = make_job(workers=8, mpiImplementation=impl)
    job["spec"]["runPolicy"] = {"suspend": True}
    client.seed(MPIJOBS, job)
    ctrl.sync("default", "test")
    # service/configmap/secret exist; launcher created suspended; NO workers
    assert client.services.get("default", "test")
    assert client.configmaps.get("default", "test-config")
    assert client.secrets.get("default", "test-ssh")
    launcher = client.jobs.get("default", "test-launcher")
    assert launcher["spec"]["suspend"] is True
    assert client.pods.list("default", builders.worker_selector("test")) == []
    st = client.mpijobs.get("default", "test")["status"]
    assert t.get_condition(st, c.JOB_CREATED)["status"] == "True"
    assert t.get_condition(st, c.JOB_SUSPENDED)["status"] == "True"
    running = t.get_condition(st, c.JOB_RUNNING)
    assert running is not None and running["status"] == "False"
    # both replica-status maps initialized (reference expects empty structs)
    assert "Launcher" in st["replicaStatuses"]
    assert "Worker" in st["replicaStatuses"]
    assert st.get("startTime") is None


# ---- TestSuspendedRunningMPIJob (:1010) ----
def test_suspend_running_job_full_semantics():
    client, ctrl = make_controller()
    client.seed(MPIJOBS, make_job(workers=8))
    ctrl.sync("default", "test")
    for i in range(8):
        p = client.pods.get("default", f"test-worker-{i}")
        p["status"] = {"phase": "Running"}
        client.pods.update("default", p)
    ctrl.sync("default", "test")
    st = client.mpijobs.get("default", "test")["status"]
    assert st["replicaStatuses"]["Worker"]["active"] == 8

    job = client.mpijobs.get("default", "test")
    job["spec"]["runPolicy"] = {"suspend": True}
    client.mpijobs.update("default", job)
    ctrl.sync("default", "test")
    # launcher suspended in place, all workers deleted
    assert client.jobs.get("default", "test-launcher")["spec"]["suspend"] is True
    assert client.pods.list("default", builders.worker_selector("test")) == []
    st = client.mpijobs.get("default", "test")["status"]
    assert t.get_condition(st, c.JOB_SUSPENDED)["status"] == "True"
    running = t.get_condition(st, c.JOB_RUNNING)
    assert running is not None and running["status"] == "False"
    assert st["replicaStatuses"]["Worker"].get("active", 0) == 0


# ---- TestResumeMPIJobWithExistingLauncher (:1207): KEP-2926 sync ----
def test_resume_syncs_kueue_scheduling_directives():
    client, ctrl = make_controller()
    job = make_job(workers=2)
    job["spec"]["runPolicy"] = {"suspend": True}
    client.seed(MPIJOBS, job)
    ctrl.sync("default", "test")
    assert client.jobs.get("default", "test-launcher")["spec"]["suspend"] is True

    # Kueue admits the job: injects scheduling directives into the MPIJob's
    # launcher template AFTER the launcher Job already exists, then resumes
    job = client.mpijobs.get("default", "test")
    tmpl = job["spec"]["mpiReplicaSpecs"]["Launcher"]["template"]
    tmpl["spec"]["nodeSelector"] = {"foo": "bar"}
    tmpl["spec"]["tolerations"] = [
        {"key": "gpu", "operator": "Equal", "value": "true",
         "effect": "NoSchedule"}]
    tmpl["spec"]["schedulingGates"] = [{"name": "kueue.x-k8s.io/topology"}]
    tmpl.setdefault("metadata", {}).setdefault("annotations", {})[
        "kueue.x-k8s.io/workload"] = "my-workload"
    job["spec"]["runPolicy"]["suspend"] = False
    client.mpijobs.update("default", job)
    ctrl.sync("default", "test")

    launcher = client.jobs.get("default", "test-launcher")
    assert launcher["spec"]["suspend"] is False
    lt = launcher["spec"]["template"]
    assert lt["spec"]["nodeSelector"] == {"foo": "bar"}
    assert lt["spec"]["tolerations"][0]["key"] == "gpu"
    assert lt["spec"]["schedulingGates"][0]["name"] == "kueue.x-k8s.io/topology"
    # workers created on resume
    assert len(client.pods.list("default", builders.worker_selector("test"))) == 2
    st = client.mpijobs.get("default", "test")["status"]
    cond = t.get_condition(st, c.JOB_SUSPENDED)
    assert cond["status"] == "False" and cond["reason"] == "MPIJobResumed"
    assert st.get("startTime") is not None


# ---- TestUnsuspendLauncherUpdateFailureDoesNotPoisonCache (:1163) ----
def test_unsuspend_launcher_update_failure_does_not_poison_state():
    """A throttled launcher Update must propagate as a sync error (so the
    rate-limited requeue retries) and must NOT leave the stored launcher
    half-mutated. (The Go regression guards the shared informer cache; the
    Python fake deep-copies on get, so the equivalent guarantee is that a
    rejected update leaves the store untouched and the error surfaces.)"""
    client, ctrl = make_controller()
    job = make_job(workers=1)
    job["spec"]["runPolicy"] = {"suspend": True}
    client.seed(MPIJOBS, job)
    ctrl.sync("default", "test")

    def throttle(verb, gvr, ns, name, obj):
        if verb == "update" and gvr.resource == "jobs":
            return True, ApiError(503, "throttled")
        return False, None

    client.reactors.append(throttle)
    job = client.mpijobs.get("default", "test")
    job["spec"]["runPolicy"]["suspend"] = False
    client.mpijobs.update("default", job)
    with pytest.raises(ApiError):
        ctrl.sync("default", "test")
    client.reactors.clear()
    # stored launcher NOT mutated by the failed update
    assert client.jobs.get("default", "test-launcher")["spec"]["suspend"] is True


# ---- replica-status matrix (:1415/:1468) ----
def _running_launcher_pod(client):
    launcher = client.jobs.get("default", "test-launcher")
    launcher["spec"].setdefault("selector", {"matchLabels": {"controller-uid": "uid1"}})
    client.jobs.update("default", launcher)
    client.pods.create("default", {
        "metadata": {"name": "test-launcher-pod", "namespace": "default",
                     "labels": {"controller-uid": "uid1"}},
        "status": {"phase": "Running"}})


def test_launcher_active_worker_not_ready():
    client, ctrl = make_controller()
    client.seed(MPIJOBS, make_job(workers=8))
    ctrl.sync("default", "test")
    _running_launcher_pod(client)
    # workers stay Pending
    for i in range(8):
        p = client.pods.get("default", f"test-worker-{i}")
        p["status"] = {"phase": "Pending"}
        client.pods.update("default", p)
    ctrl.sync("default", "test")
    st = client.mpijobs.get("default", "test")["status"]
    assert st["replicaStatuses"]["Launcher"]["active"] == 1
    assert st["replicaStatuses"]["Worker"].get("active", 0) == 0
    # not all ranks can run: no Running=True condition yet
    running = t.get_condition(st, c.JOB_RUNNING)
    assert running is None or running["status"] != "True"


def test_launcher_active_worker_ready_sets_running():
    client, ctrl = make_controller()
    client.seed(MPIJOBS, make_job(workers=8))
    ctrl.sync("default", "test")
    _running_launcher_pod(client)
    for i in range(8):
        p = client.pods.get("default", f"test-worker-{i}")
        p["status"] = {"phase": "Running"}
        client.pods.update("default", p)
    ctrl.sync("default", "test")
    st = client.mpijobs.get("default", "test")["status"]
    assert st["replicaStatuses"]["Launcher"]["active"] == 1
    assert st["replicaStatuses"]["Worker"]["active"] == 8
    assert t.get_condition(st, c.JOB_RUNNING)["status"] == "True"


def test_worker_evicted_fails_job():
    """Evicted worker counting → MPIJob Failed (reference :1145-1164)."""
    client, ctrl = make_controller()
    client.seed(MPIJOBS, make_job(workers=2))
    ctrl.sync("default", "test")
    p = client.pods.get("default", "test-worker-1")
    p["status"] = {"phase": "Failed", "reason": "Evicted"}
    client.pods.update("default", p)
    ctrl.sync("default", "test")
    st = client.mpijobs.get("default", "test")["status"]
    assert st["replicaStatuses"]["Worker"]["failed"] == 1
    assert t.has_condition_true(st, c.JOB_FAILED)


# ---- podgroup minResources cases (podgroup_test.go:442) ----
def _job_with_resources(launcher_req, worker_req, workers=2, sched=None):
    job = make_job(workers=workers)
    if sched is not None:
        job["spec"]["runPolicy"] = {"schedulingPolicy": sched}
    job["spec"]["mpiReplicaSpecs"]["Launcher"]["template"]["spec"][
        "containers"][0]["resources"] = {"requests": launcher_req}
    job["spec"]["mpiReplicaSpecs"]["Worker"]["template"]["spec"][
        "containers"][0]["resources"] = {"requests": worker_req}
    return job


def _pg_ctrl(priority_classes=None):
    from mpi_operator_amd.controller.podgroup import VolcanoCtrl
    return VolcanoCtrl(FakeKubeClient(), priority_classes=priority_classes)


def _defaulted(job):
    from mpi_operator_amd.controller.api import defaults
    defaults.set_defaults_mpijob(job)
    return job


def test_pg_min_resources_explicit_wins():
    job = _defaulted(make_job())
    job["spec"]["runPolicy"] = {"schedulingPolicy": {
        "minResources": {"cpu": "7", "memory": "10Gi"}}}
    got = _pg_ctrl().calculate_pg_min_resources(3, job)
    assert got == {"cpu": "7", "memory": "10Gi"}


def test_pg_min_resources_no_scheduling_policy_sums_all():
    job = _defaulted(_job_with_resources({"cpu": "2", "memory": "1Gi"},
                                         {"cpu": "10", "memory": "32Gi"},
                                         workers=2))
    got = _pg_ctrl().calculate_pg_min_resources(3, job)
    assert got["cpu"] == "22"           # 2 + 2*10
    assert got["memory"] == "65Gi"      # 1 + 2*32


def test_pg_min_resources_launcher_only():
    job = make_job(workers=2)
    del job["spec"]["mpiReplicaSpecs"]["Worker"]
    job["spec"]["mpiReplicaSpecs"]["Launcher"]["template"]["spec"][
        "containers"][0]["resources"] = {"requests": {"cpu": "2", "memory": "1Gi"}}
    got = _pg_ctrl().calculate_pg_min_resources(1, _defaulted(job))
    assert got["cpu"] == "2" and got["memory"] == "1Gi"


def test_pg_min_resources_priority_order_truncates_lower_class():
    """With worker priority > launcher priority and minMember=2, the
    reference keeps the HIGHER class in full and truncates the lower one to
    minMember-1 (podgroup.go:358-376: order[1].Replicas = minMember-1), so
    2 workers + 1 launcher are counted."""
    job = _job_with_resources({"cpu": "2", "memory": "1Gi"},
                              {"cpu": "10", "memory": "32Gi"}, workers=2)
    job["spec"]["mpiReplicaSpecs"]["Launcher"]["template"]["spec"][
        "priorityClassName"] = "low"
    job["spec"]["mpiReplicaSpecs"]["Worker"]["template"]["spec"][
        "priorityClassName"] = "high"
    got = _pg_ctrl({"low": 10, "high": 100}).calculate_pg_min_resources(
        2, _defaulted(job))
    assert got["cpu"] == "22"           # 2 workers (full) + 1 launcher
    assert got["memory"] == "65Gi"


def test_pg_min_resources_equal_priority_launcher_first():
    """Equal priorities: the worker replica count is truncated to
    minMember-1 (reference podgroup.go:362-374), so minMember=2 counts
    launcher + 1 worker."""
    job = _defaulted(_job_with_resources({"cpu": "2", "memory": "1Gi"},
                                         {"cpu": "10", "memory": "32Gi"},
                                         workers=2))
    got = _pg_ctrl().calculate_pg_min_resources(2, job)
    assert got["cpu"] == "12"
    assert got["memory"] == "33Gi"


# ---- TestResumeMPIJobWithExistingLauncher (:1207): resume must UNSUSPEND
# the existing launcher Job in place (not delete/recreate) and recreate
# the worker pods ----
def test_resume_with_existing_launcher_unsuspends_in_place():
    client, ctrl = make_controller()
    job = make_job(workers=2)
    job["spec"]["runPolicy"] = {"suspend": True}
    client.seed(MPIJOBS, job)
    ctrl.sync("default", "test")
    launcher = client.jobs.get("default", "test-launcher")
    assert launcher["spec"]["suspend"] is True
    launcher_uid = launcher["metadata"].get("uid", id(launcher))

    # resume: same launcher object gets suspend=False; workers appear
    j = client.mpijobs.get("default", "test")
    j["spec"]["runPolicy"]["suspend"] = False
    client.mpijobs.update("default", j)
    ctrl.sync("default", "test")
    launcher2 = client.jobs.get("default", "test-launcher")
    assert launcher2["spec"]["suspend"] is False
    assert launcher2["metadata"].get("uid", id(launcher2)) == launcher_uid
    workers = client.pods.list("default", builders.worker_selector("test"))
    assert len(workers) == 2
    st = client.mpijobs.get("default", "test")["status"]
    assert t.get_condition(st, c.JOB_SUSPENDED)["status"] == "False"


# ---- TestLauncherSucceededWithRunningPod (:681): the batch Job's Complete
# condition is AUTHORITATIVE — a launcher pod still observed Running must
# not keep the job Active or block the Succeeded transition ----
def test_launcher_succeeded_while_pod_still_running():
    from tests.test_controller import _complete_launcher
    client, ctrl = make_controller()
    client.seed(MPIJOBS, make_job(workers=1))
    ctrl.sync("default", "test")
    launcher = client.jobs.get("default", "test-launcher")
    launcher["spec"]["selector"] = {"matchLabels": {"controller-uid": "lr1"}}
    client.jobs.update("default", launcher)
    # lagging pod observation: still Running after the Job completed
    client.pods.create("default", {
        "metadata": {"name": "test-launcher-pod", "namespace": "default",
                     "labels": {"controller-uid": "lr1"}},
        "status": {"phase": "Running"}})
    _complete_launcher(client, succeeded=True)
    ctrl.sync("default", "test")
    stt = client.mpijobs.get("default", "test")["status"]
    assert t.has_condition_true(stt, c.JOB_SUCCEEDED)
    running = t.get_condition(stt, c.JOB_RUNNING)
    assert running is None or running["status"] == "False"
    ls = stt["replicaStatuses"]["Launcher"]
    assert ls["succeeded"] == 1
    assert ls.get("active", 0) == 0
