"""Client interface + group/version/resource descriptors."""
from __future__ import annotations

from dataclasses import dataclass
from typing import Iterable, Optional


class ApiError(Exception):
    def __init__(self, code: int, msg: str = ""):
        super().__init__(f"api error {code}: {msg}")
        self.code = code


class NotFound(ApiError):
    def __init__(self, msg: str = ""):
        super().__init__(404, msg)


class Conflict(ApiError):
    def __init__(self, msg: str = ""):
        super().__init__(409, msg)


@dataclass(frozen=True)
class GVR:
    group: str  # "" for core
    version: str
    resource: str  # plural
    kind: str
    namespaced: bool = True

    @property
    def api_version(self) -> str:
        return f"{self.group}/{self.version}" if self.group else self.version


PODS = GVR("", "v1", "pods", "Pod")
SERVICES = GVR("", "v1", "services", "Service")
CONFIGMAPS = GVR("", "v1", "configmaps", "ConfigMap")
SECRETS = GVR("", "v1", "secrets", "Secret")
EVENTS = GVR("", "v1", "events", "Event")
JOBS = GVR("batch", "v1", "jobs", "Job")
LEASES = GVR("coordination.k8s.io", "v1", "leases", "Lease")
MPIJOBS = GVR("kubeflow.org", "v2beta1", "mpijobs", "MPIJob")
PODGROUPS_VOLCANO = GVR("scheduling.volcano.sh", "v1beta1", "podgroups", "PodGroup")
PODGROUPS_SCHED = GVR("scheduling.x-k8s.io", "v1alpha1", "podgroups", "PodGroup")


class ResourceClient:
    """Typed-by-kind CRUD over dict objects."""

    def get(self, namespace: str, name: str) -> dict:
        raise NotImplementedError

    def list(self, namespace: str, label_selector: Optional[dict] = None) -> list[dict]:
        raise NotImplementedError

    def create(self, namespace: str, obj: dict) -> dict:
        raise NotImplementedError

    def update(self, namespace: str, obj: dict) -> dict:
        raise NotImplementedError

    def update_status(self, namespace: str, obj: dict) -> dict:
        raise NotImplementedError

    def delete(self, namespace: str, name: str) -> None:
        raise NotImplementedError


class KubeClient:
    """Bundle of per-kind clients; implementations provide .resource(gvr)."""

    def resource(self, gvr: GVR) -> ResourceClient:
        raise NotImplementedError

    @property
    def pods(self) -> ResourceClient:
        return self.resource(PODS)

    @property
    def services(self) -> ResourceClient:
        return self.resource(SERVICES)

    @property
    def configmaps(self) -> ResourceClient:
        return self.resource(CONFIGMAPS)

    @property
    def secrets(self) -> ResourceClient:
        return self.resource(SECRETS)

    @property
    def jobs(self) -> ResourceClient:
        return self.resource(JOBS)

    @property
    def events(self) -> ResourceClient:
        return self.resource(EVENTS)

    @property
    def leases(self) -> ResourceClient:
        return self.resource(LEASES)

    @property
    def mpijobs(self) -> ResourceClient:
        return self.resource(MPIJOBS)


def match_labels(obj: dict, selector: Optional[dict]) -> bool:
    if not selector:
        return True
    labels = obj.get("metadata", {}).get("labels", {}) or {}
    return all(labels.get(k) == v for k, v in selector.items())
