from .base import SdkModel  # noqa: F401
from .v2beta1_models import (  # noqa: F401
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
