"""API defaulting/validation + podgroup math tests — mirrors the reference's
default_test.go / validation_test.go / podgroup_test.go matrices."""
import pytest

from mpi_operator_amd.controller.api import constants as c
from mpi_operator_amd.controller.api.defaults import set_defaults_mpijob
from mpi_operator_amd.controller.api.validation import validate_mpijob
from mpi_operator_amd.controller.podgroup import (SchedulerPluginsCtrl,
                                                  VolcanoCtrl,
                                                  calculate_min_available,
                                                  calculate_priority_class_name,
                                                  cal_pg_min_resource,
                                                  parse_quantity)
from mpi_operator_amd.controller.client import FakeKubeClient

from test_controller import make_job


def test_set_defaults():
    job = {"metadata": {"name": "x"},
           "spec": {"mpiReplicaSpecs": {
               "Launcher": {"template": {"spec": {"containers": [{}]}}},
               "Worker": {"replicas": 3, "template": {"spec": {"containers": [{}]}}}}}}
    set_defaults_mpijob(job)
    s = job["spec"]
    assert s["slotsPerWorker"] == 1
    assert s["sshAuthMountPath"] == "/root/.ssh"
    assert s["mpiImplementation"] == "OpenMPI"
    assert s["launcherCreationPolicy"] == "AtStartup"
    assert s["runPolicy"]["cleanPodPolicy"] == "None"
    assert s["mpiReplicaSpecs"]["Launcher"]["replicas"] == 1
    assert s["mpiReplicaSpecs"]["Launcher"]["restartPolicy"] == "OnFailure"
    assert s["mpiReplicaSpecs"]["Worker"]["restartPolicy"] == "Never"


@pytest.mark.parametrize("mutate,frag", [
    (lambda j: j["spec"].pop("slotsPerWorker"), "slotsPerWorker"),
    (lambda j: j["spec"].pop("sshAuthMountPath"), "sshAuthMountPath"),
    (lambda j: j["spec"].update(mpiImplementation="Unknown"), "mpiImplementation"),
    (lambda j: j["spec"]["runPolicy"].pop("cleanPodPolicy"), "cleanPodPolicy"),
    (lambda j: j["spec"]["runPolicy"].update(ttlSecondsAfterFinished=-1), "ttlSecondsAfterFinished"),
    (lambda j: j["spec"]["runPolicy"].update(managedBy="unknown/ctrl"), "managedBy"),
    (lambda j: j["spec"]["mpiReplicaSpecs"].pop("Launcher"), "Launcher"),
    (lambda j: j["spec"]["mpiReplicaSpecs"]["Launcher"].update(replicas=2), "must be 1"),
    (lambda j: j["spec"]["mpiReplicaSpecs"]["Worker"].update(replicas=0), "greater than or equal to 1"),
    (lambda j: j["spec"]["mpiReplicaSpecs"]["Worker"].update(restartPolicy="Always"), "restartPolicy"),
    (lambda j: j["spec"]["mpiReplicaSpecs"]["Worker"]["template"]["spec"].update(containers=[]), "containers"),
    (lambda j: j["metadata"].update(name="1-bad-name"), "DNS"),
])
def test_validation_rejects(mutate, frag):
    job = make_job()
    set_defaults_mpijob(job)
    mutate(job)
    errs = validate_mpijob(job)
    assert errs and any(frag in e for e in errs), errs


def test_validation_accepts_valid():
    job = make_job()
    set_defaults_mpijob(job)
    assert validate_mpijob(job) == []


def test_min_available():
    job = make_job(workers=4)
    assert calculate_min_available(job) == 5
    job["spec"]["runPolicy"] = {"schedulingPolicy": {"minAvailable": 3}}
    assert calculate_min_available(job) == 3


def test_priority_class_precedence():
    job = make_job()
    assert calculate_priority_class_name(job) == ""
    job["spec"]["mpiReplicaSpecs"]["Worker"]["template"]["spec"]["priorityClassName"] = "w"
    assert calculate_priority_class_name(job) == "w"
    job["spec"]["mpiReplicaSpecs"]["Launcher"]["template"]["spec"]["priorityClassName"] = "l"
    assert calculate_priority_class_name(job) == "l"
    job["spec"]["runPolicy"] = {"schedulingPolicy": {"priorityClass": "sp"}}
    assert calculate_priority_class_name(job) == "sp"


def test_pg_min_resources_counts_first_min_member():
    # 4 workers, minMember 3 → launcher + 2 workers counted (workers lose ties)
    job = make_job(workers=4)
    job["spec"]["mpiReplicaSpecs"]["Launcher"]["replicas"] = 1
    job["spec"]["mpiReplicaSpecs"]["Launcher"]["template"]["spec"]["containers"][0][
        "resources"] = {"requests": {"cpu": "1"}}
    mr = cal_pg_min_resource(3, job, None)
    # 2 workers * 8 GPUs + 1 launcher cpu
    assert mr["amd.com/gpu"] == "16"
    assert mr["cpu"] == "1"


def test_quantities():
    assert parse_quantity("100m") * 10 == 1
    assert parse_quantity("1Gi") == 2**30
    assert parse_quantity("2") == 2


def test_volcano_podgroup_shape():
    client = FakeKubeClient()
    v = VolcanoCtrl(client)
    job = make_job(workers=2)
    job["metadata"]["uid"] = "u1"
    job["metadata"]["annotations"] = {"scheduling.volcano.sh/queue-name": "q1"}
    pg = v.new_pod_group(job)
    assert pg["spec"]["minMember"] == 3
    assert pg["spec"]["queue"] == "q1"
    assert pg["spec"]["minResources"]["amd.com/gpu"] == "16"
    tmpl = {"spec": {"containers": [{}]}}
    v.decorate_pod_template(tmpl, "test")
    assert tmpl["spec"]["schedulerName"] == "volcano"
    assert tmpl["metadata"]["annotations"]["scheduling.k8s.io/group-name"] == "test"


def test_sched_plugins_podgroup_shape():
    client = FakeKubeClient()
    s = SchedulerPluginsCtrl(client, scheduler_name="coscheduler")
    job = make_job(workers=2)
    job["spec"]["runPolicy"] = {"schedulingPolicy": {"scheduleTimeoutSeconds": 120}}
    pg = s.new_pod_group(job)
    assert pg["spec"]["minMember"] == 3
    assert pg["spec"]["scheduleTimeoutSeconds"] == 120
    tmpl = {"spec": {"containers": [{}]}}
    s.decorate_pod_template(tmpl, "test")
    assert tmpl["spec"]["schedulerName"] == "coscheduler"
    assert tmpl["metadata"]["labels"]["scheduling.x-k8s.io/pod-group"] == "test"


def test_gang_scheduling_creates_podgroup_and_decorates_pods():
    from mpi_operator_amd.controller import MPIJobController
    from test_controller import fake_keygen
    client = FakeKubeClient()
    ctrl = MPIJobController(client, podgroup_ctrl=VolcanoCtrl(client),
                            keygen=fake_keygen)
    job = make_job(workers=2)
    client.seed(__import__("mpi_operator_amd.controller.client.base",
                           fromlist=["MPIJOBS"]).MPIJOBS, job)
    ctrl.sync("default", "test")
    pg = client.resource(VolcanoCtrl.gvr).get("default", "test")
    assert pg["spec"]["minMember"] == 3
    pod = client.pods.get("default", "test-worker-0")
    assert pod["spec"]["schedulerName"] == "volcano"
    assert pod["metadata"]["annotations"]["scheduling.k8s.io/group-name"] == "test"
    launcher = client.jobs.get("default", "test-launcher")
    assert launcher["spec"]["template"]["spec"]["schedulerName"] == "volcano"
