"""SDK model tests — serialization round-trips (the role of the reference's
generated sdk/python/v2beta1/test/test_v2beta1_*.py) plus a cross-check that
SDK output is accepted by the controller's defaulting + validation."""
import os
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(REPO, "sdk", "python", "v2beta1"))

from mpijob import (V2beta1JobCondition, V2beta1JobStatus, V2beta1MPIJob,  # noqa: E402
                    V2beta1MPIJobSpec, V2beta1ReplicaSpec, V2beta1ReplicaStatus,
                    V2beta1RunPolicy, V2beta1SchedulingPolicy)


def sample_job():
    return V2beta1MPIJob(
        api_version="kubeflow.org/v2beta1",
        kind="MPIJob",
        metadata={"name": "demo", "namespace": "ns"},
        spec=V2beta1MPIJobSpec(
            slots_per_worker=8,
            mpi_implementation="OpenMPI",
            run_policy=V2beta1RunPolicy(
                clean_pod_policy="Running",
                backoff_limit=3,
                scheduling_policy=V2beta1SchedulingPolicy(min_available=2, queue="q"),
            ),
            mpi_replica_specs={
                "Launcher": V2beta1ReplicaSpec(replicas=1, template={
                    "spec": {"containers": [{"name": "l", "image": "img"}]}}),
                "Worker": V2beta1ReplicaSpec(replicas=2, template={
                    "spec": {"containers": [{"name": "w", "image": "img"}]}}),
            },
        ),
    )


def test_to_dict_camel_case_wire_format():
    d = sample_job().to_dict()
    assert d["spec"]["slotsPerWorker"] == 8
    assert d["spec"]["runPolicy"]["cleanPodPolicy"] == "Running"
    assert d["spec"]["runPolicy"]["schedulingPolicy"]["minAvailable"] == 2
    assert d["spec"]["mpiReplicaSpecs"]["Worker"]["replicas"] == 2
    assert "slots_per_worker" not in d["spec"]


def test_round_trip_equality():
    job = sample_job()
    back = V2beta1MPIJob.from_dict(job.to_dict())
    assert back == job
    assert isinstance(back.spec, V2beta1MPIJobSpec)
    assert isinstance(back.spec.run_policy.scheduling_policy, V2beta1SchedulingPolicy)
    assert isinstance(back.spec.mpi_replica_specs["Worker"], V2beta1ReplicaSpec)


def test_status_models_round_trip():
    st = V2beta1JobStatus(
        conditions=[V2beta1JobCondition(type="Running", status="True", reason="r")],
        replica_statuses={"Worker": V2beta1ReplicaStatus(active=2, failed=0)},
        start_time="2026-01-01T00:00:00Z",
    )
    back = V2beta1JobStatus.from_dict(st.to_dict())
    assert back == st
    assert back.conditions[0].type == "Running"
    assert back.replica_statuses["Worker"].active == 2


def test_unknown_kwarg_rejected():
    with pytest.raises(TypeError):
        V2beta1RunPolicy(bogus=1)


def test_none_fields_omitted_from_wire():
    d = V2beta1RunPolicy(backoff_limit=None, suspend=False).to_dict()
    assert "backoffLimit" not in d
    assert d == {"suspend": False}


def test_sdk_output_passes_controller_validation():
    sys.path.insert(0, REPO)
    from mpi_operator_amd.controller.api import defaults, validation

    obj = sample_job().to_dict()
    defaults.set_defaults_mpijob(obj)
    errs = validation.validate_mpijob(obj)
    assert errs == [], errs
