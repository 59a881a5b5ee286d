"""API defaulting/validation + podgroup math tests — mirrors the reference's
default_test.go / validation_test.go / podgroup_test.go matrices."""
import os
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


# ---- CRD schema round-trip (hack/gen_crd.py) ----

def _validate_schema(obj, schema, path="$"):
    """Minimal OpenAPI v3 structural validator (enough for the CRD test:
    type/enum/required/properties/items/additionalProperties)."""
    t_ = schema.get("type")
    errs = []
    if "anyOf" in schema:
        if not any(not _validate_schema(obj, s, path) for s in schema["anyOf"]):
            errs.append(f"{path}: matches no anyOf branch")
        return errs
    if t_ == "object":
        if not isinstance(obj, dict):
            return [f"{path}: expected object, got {type(obj).__name__}"]
        for req in schema.get("required", []):
            if req not in obj:
                errs.append(f"{path}: missing required {req}")
        props = schema.get("properties", {})
        addl = schema.get("additionalProperties")
        preserve = schema.get("x-kubernetes-preserve-unknown-fields")
        for k, v in obj.items():
            if k in props:
                errs += _validate_schema(v, props[k], f"{path}.{k}")
            elif isinstance(addl, dict):
                errs += _validate_schema(v, addl, f"{path}.{k}")
            elif not preserve and props and addl is None:
                errs.append(f"{path}: unknown field {k}")
    elif t_ == "array":
        if not isinstance(obj, list):
            return [f"{path}: expected array"]
        for i, v in enumerate(obj):
            errs += _validate_schema(v, schema.get("items", {}), f"{path}[{i}]")
    elif t_ == "string":
        if not isinstance(obj, str):
            return [f"{path}: expected string, got {obj!r}"]
        if "enum" in schema and obj not in schema["enum"]:
            errs.append(f"{path}: {obj!r} not in {schema['enum']}")
    elif t_ == "integer":
        if not isinstance(obj, int) or isinstance(obj, bool):
            return [f"{path}: expected integer, got {obj!r}"]
        if "minimum" in schema and obj < schema["minimum"]:
            errs.append(f"{path}: {obj} < minimum {schema['minimum']}")
    elif t_ == "boolean":
        if not isinstance(obj, bool):
            return [f"{path}: expected boolean, got {obj!r}"]
    return errs


def _crd_schema():
    import yaml
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    with open(os.path.join(root, "manifests", "base",
                           "kubeflow.org_mpijobs.yaml")) as f:
        crd = yaml.safe_load(f)
    return crd["spec"]["versions"][0]["schema"]["openAPIV3Schema"]


def test_crd_generator_in_sync():
    import subprocess, sys
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run([sys.executable, os.path.join(root, "hack", "gen_crd.py"),
                        "--check"], capture_output=True, text=True)
    assert r.returncode == 0, r.stderr


def test_example_yamls_validate_against_crd_schema():
    import glob
    import yaml
    schema = _crd_schema()
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    files = glob.glob(os.path.join(root, "examples", "v2beta1", "**", "*.yaml"),
                      recursive=True)
    checked = 0
    for f in files:
        with open(f) as fh:
            for doc in yaml.safe_load_all(fh):
                if not doc or doc.get("kind") != "MPIJob":
                    continue
                errs = _validate_schema(doc, schema)
                assert not errs, f"{f}: {errs}"
                checked += 1
    assert checked >= 3  # pi, resnet, bert examples at minimum


def test_crd_schema_rejects_malformed():
    schema = _crd_schema()
    bad = {"apiVersion": "kubeflow.org/v2beta1", "kind": "MPIJob",
           "metadata": {"name": "x"},
           "spec": {"mpiImplementation": "NotReal",
                    "slotsPerWorker": 0,
                    "mpiReplicaSpecs": {"Worker": {
                        "replicas": "two",
                        "template": {"spec": {"containers": [
                            {"image": "x"}]}}}}}}
    errs = _validate_schema(bad, schema)
    joined = "\n".join(errs)
    assert "NotReal" in joined          # bad enum
    assert "slotsPerWorker: 0" in joined or "minimum" in joined
    assert "expected integer" in joined  # replicas: "two"
    assert "missing required name" in joined  # container without name
