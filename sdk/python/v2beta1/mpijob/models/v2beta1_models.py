"""MPIJob v2beta1 SDK models — field-for-field the reference SDK's models
(reference sdk/python/v2beta1/mpijob/models/v2beta1_*.py; wire schema =
pkg/apis/kubeflow/v2beta1/types.go there). Pod templates and metadata are
plain dicts (or `kubernetes` client objects, which serialize themselves)."""
from __future__ import annotations

from .base import SdkModel


class V2beta1SchedulingPolicy(SdkModel):
    attribute_map = {
        "min_available": "minAvailable",
        "queue": "queue",
        "min_resources": "minResources",
        "priority_class": "priorityClass",
        "schedule_timeout_seconds": "scheduleTimeoutSeconds",
    }
    openapi_types = {
        "min_available": int,
        "queue": str,
        "min_resources": dict,
        "priority_class": str,
        "schedule_timeout_seconds": int,
    }


class V2beta1RunPolicy(SdkModel):
    attribute_map = {
        "clean_pod_policy": "cleanPodPolicy",
        "ttl_seconds_after_finished": "ttlSecondsAfterFinished",
        "active_deadline_seconds": "activeDeadlineSeconds",
        "backoff_limit": "backoffLimit",
        "scheduling_policy": "schedulingPolicy",
        "suspend": "suspend",
        "managed_by": "managedBy",
    }
    openapi_types = {
        "clean_pod_policy": str,
        "ttl_seconds_after_finished": int,
        "active_deadline_seconds": int,
        "backoff_limit": int,
        "scheduling_policy": V2beta1SchedulingPolicy,
        "suspend": bool,
        "managed_by": str,
    }


class V2beta1ReplicaSpec(SdkModel):
    attribute_map = {
        "replicas": "replicas",
        "template": "template",
        "restart_policy": "restartPolicy",
    }
    openapi_types = {
        "replicas": int,
        "template": dict,
        "restart_policy": str,
    }


class V2beta1MPIJobSpec(SdkModel):
    attribute_map = {
        "slots_per_worker": "slotsPerWorker",
        "run_launcher_as_worker": "runLauncherAsWorker",
        "run_policy": "runPolicy",
        "mpi_replica_specs": "mpiReplicaSpecs",
        "ssh_auth_mount_path": "sshAuthMountPath",
        "launcher_creation_policy": "launcherCreationPolicy",
        "mpi_implementation": "mpiImplementation",
    }
    openapi_types = {
        "slots_per_worker": int,
        "run_launcher_as_worker": bool,
        "run_policy": V2beta1RunPolicy,
        "mpi_replica_specs": ["dict", V2beta1ReplicaSpec],
        "ssh_auth_mount_path": str,
        "launcher_creation_policy": str,
        "mpi_implementation": str,
    }


class V2beta1JobCondition(SdkModel):
    attribute_map = {
        "type": "type",
        "status": "status",
        "reason": "reason",
        "message": "message",
        "last_update_time": "lastUpdateTime",
        "last_transition_time": "lastTransitionTime",
    }
    openapi_types = {k: str for k in attribute_map}


class V2beta1ReplicaStatus(SdkModel):
    attribute_map = {
        "active": "active",
        "succeeded": "succeeded",
        "failed": "failed",
        "selector": "selector",
    }
    openapi_types = {"active": int, "succeeded": int, "failed": int, "selector": str}


class V2beta1JobStatus(SdkModel):
    attribute_map = {
        "conditions": "conditions",
        "replica_statuses": "replicaStatuses",
        "start_time": "startTime",
        "completion_time": "completionTime",
        "last_reconcile_time": "lastReconcileTime",
    }
    openapi_types = {
        "conditions": ["list", V2beta1JobCondition],
        "replica_statuses": ["dict", V2beta1ReplicaStatus],
        "start_time": str,
        "completion_time": str,
        "last_reconcile_time": str,
    }


class V2beta1MPIJob(SdkModel):
    attribute_map = {
        "api_version": "apiVersion",
        "kind": "kind",
        "metadata": "metadata",
        "spec": "spec",
        "status": "status",
    }
    openapi_types = {
        "api_version": str,
        "kind": str,
        "metadata": dict,
        "spec": V2beta1MPIJobSpec,
        "status": V2beta1JobStatus,
    }


class V2beta1MPIJobList(SdkModel):
    attribute_map = {
        "api_version": "apiVersion",
        "kind": "kind",
        "metadata": "metadata",
        "items": "items",
    }
    openapi_types = {
        "api_version": str,
        "kind": str,
        "metadata": dict,
        "items": ["list", V2beta1MPIJob],
    }
