"""Kubernetes client layer: a typed-by-kind interface over dict objects with
two implementations — an in-memory fake (unit/integration tests, the analog
of the reference's fake clientsets, reference
pkg/controller/mpi_job_controller_test.go:70-110) and a REST client
(requests; kubeconfig or in-cluster)."""
from .base import KubeClient, ResourceClient, Conflict, NotFound, ApiError, GVR
from .fake import FakeKubeClient

__all__ = ["KubeClient", "ResourceClient", "FakeKubeClient",
           "Conflict", "NotFound", "ApiError", "GVR"]
