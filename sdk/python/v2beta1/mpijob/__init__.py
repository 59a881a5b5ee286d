"""kubeflow-mpi SDK (v2beta1) — client models for the MPIJob CRD.

Same public surface as the reference's OpenAPI-generated SDK
(reference sdk/python/v2beta1/mpijob/models/): V2beta1MPIJob and friends,
each with attribute_map-driven to_dict()/from_dict() producing exactly the
camelCase JSON the apiserver expects, for use with the `kubernetes`
client's CustomObjectsApi (see ../tensorflow-mnist.py there). Hand-written
instead of generated — the models are thin and stable."""
from .models import (  # noqa: F401
    V2beta1JobCondition,
    V2beta1JobStatus,
    V2beta1MPIJob,
    V2beta1MPIJobList,
    V2beta1MPIJobSpec,
    V2beta1ReplicaSpec,
    V2beta1ReplicaStatus,
    V2beta1RunPolicy,
    V2beta1SchedulingPolicy,
)

__version__ = "2.0.0-amd"

__all__ = [
    "V2beta1JobCondition", "V2beta1JobStatus", "V2beta1MPIJob",
    "V2beta1MPIJobList", "V2beta1MPIJobSpec", "V2beta1ReplicaSpec",
    "V2beta1ReplicaStatus", "V2beta1RunPolicy", "V2beta1SchedulingPolicy",
]
