"""Gang-scheduling adapters: Volcano (scheduling.volcano.sh/v1beta1) and
scheduler-plugins (scheduling.x-k8s.io/v1alpha1) PodGroups — behavioral
parity with reference pkg/controller/podgroup.go:42-475."""
from __future__ import annotations

import re
from fractions import Fraction

from .api import constants as c
from .api import types as t
from .client.base import GVR, PODGROUPS_SCHED, PODGROUPS_VOLCANO

VOLCANO_QUEUE_ANNOTATION = "scheduling.volcano.sh/queue-name"
VOLCANO_GROUP_ANNOTATION = "scheduling.k8s.io/group-name"
SCHED_PLUGINS_POD_GROUP_LABEL = "scheduling.x-k8s.io/pod-group"

_SUFFIX = {
    "": 1, "m": Fraction(1, 1000), "k": 10**3, "M": 10**6, "G": 10**9,
    "T": 10**12, "P": 10**15, "E": 10**18,
    "Ki": 2**10, "Mi": 2**20, "Gi": 2**30, "Ti": 2**40, "Pi": 2**50, "Ei": 2**60,
}
_QTY_RE = re.compile(r"^([0-9.]+)([a-zA-Z]*)$")


def parse_quantity(q) -> Fraction:
    if isinstance(q, (int, float)):
        return Fraction(q)
    m = _QTY_RE.match(str(q))
    if not m:
        raise ValueError(f"bad quantity {q!r}")
    num, suf = m.groups()
    if suf not in _SUFFIX:
        raise ValueError(f"bad quantity suffix {q!r}")
    return Fraction(num) * _SUFFIX[suf]


def format_quantity(v: Fraction) -> str:
    if v.denominator == 1:
        n = v.numerator
        # canonicalize large byte counts back to binary suffixes, matching
        # k8s resource.Quantity String() ("1Gi"+2×"32Gi" prints "65Gi")
        if n != 0 and n % 1024 == 0:
            for suf, mult in (("Ei", 1 << 60), ("Pi", 1 << 50), ("Ti", 1 << 40),
                              ("Gi", 1 << 30), ("Mi", 1 << 20), ("Ki", 1 << 10)):
                if n % mult == 0:
                    return f"{n // mult}{suf}"
        return str(n)
    milli = v * 1000
    if milli.denominator == 1:
        return f"{milli.numerator}m"
    return str(float(v))


def add_resources(min_resources: dict, resources: dict, replicas: int) -> None:
    """reference podgroup.go:420-443 — requests, falling back to limits."""
    if not resources:
        return
    merged = dict(resources.get("requests", {}) or {})
    for name, lim in (resources.get("limits", {}) or {}).items():
        merged.setdefault(name, lim)
    for name, qty in merged.items():
        v = parse_quantity(qty) * replicas
        if name in min_resources:
            v += parse_quantity(min_resources[name])
        min_resources[name] = format_quantity(v)


def calculate_min_available(job: dict) -> int:
    sp = t.scheduling_policy(job)
    if sp and sp.get("minAvailable") is not None:
        return int(sp["minAvailable"])
    return t.worker_replicas(job) + 1


def calculate_priority_class_name(job: dict) -> str:
    sp = t.scheduling_policy(job)
    if sp and sp.get("priorityClass"):
        return sp["priorityClass"]
    for rt in (c.MPI_REPLICA_TYPE_LAUNCHER, c.MPI_REPLICA_TYPE_WORKER):
        rs = t.replica_specs(job).get(rt)
        pc = t.deep_get(rs, "template", "spec", "priorityClassName", default="")
        if pc:
            return pc
    return ""


def cal_pg_min_resource(min_member: int, job: dict, priority_classes: dict | None) -> dict:
    """reference podgroup.go:337-388: order replicas by PriorityClass value
    (workers lose ties), count only the first min_member replicas."""
    order = []
    for rt, rs in t.replica_specs(job).items():
        pc_name = t.deep_get(rs, "template", "spec", "priorityClassName", default="")
        prio = 0
        if pc_name and priority_classes and pc_name in priority_classes:
            prio = int(priority_classes[pc_name])
        order.append({"priority": prio, "type": rt,
                      "replicas": int(rs.get("replicas") or 0),
                      "template": rs.get("template", {})})
    order.sort(key=lambda r: -r["priority"])
    if not order:
        return {}
    replicas = order[0]["replicas"] + (order[1]["replicas"] if len(order) > 1 else 0)
    if replicas > min_member and len(order) > 1:
        if order[0]["priority"] == order[1]["priority"]:
            w = next((i for i, r in enumerate(order)
                      if r["type"] == c.MPI_REPLICA_TYPE_WORKER), -1)
            if w == -1:
                return {}
            order[w]["replicas"] = min_member - 1
        else:
            order[1]["replicas"] = min_member - 1
    min_resources: dict = {}
    for rp in order:
        for cont in t.deep_get(rp, "template", "spec", "containers", default=[]):
            add_resources(min_resources, cont.get("resources", {}), rp["replicas"])
    return min_resources


class PodGroupControl:
    """Interface parity: newPodGroup/decoratePodTemplateSpec/
    calculatePGMinResources/pgSpecsAreEqual (reference podgroup.go:42-65)."""

    gvr: GVR
    scheduler_name: str

    def __init__(self, kube_client, scheduler_name: str, priority_classes: dict | None = None):
        self.client = kube_client.resource(self.gvr)
        self.scheduler_name = scheduler_name
        self.priority_classes = priority_classes or {}

    def new_pod_group(self, job: dict) -> dict:
        raise NotImplementedError

    def decorate_pod_template(self, tmpl: dict, job_name: str) -> None:
        raise NotImplementedError

    def calculate_pg_min_resources(self, min_member: int, job: dict):
        sp = t.scheduling_policy(job)
        if sp and sp.get("minResources") is not None:
            return sp["minResources"]
        if min_member == 0:
            return None
        return cal_pg_min_resource(min_member, job, self.priority_classes)

    def pg_specs_equal(self, a: dict, b: dict) -> bool:
        return a.get("spec") == b.get("spec")


class VolcanoCtrl(PodGroupControl):
    gvr = PODGROUPS_VOLCANO

    def __init__(self, kube_client, priority_classes=None):
        super().__init__(kube_client, "volcano", priority_classes)

    def new_pod_group(self, job: dict) -> dict:
        min_member = calculate_min_available(job)
        queue = t.meta(job).get("annotations", {}).get(VOLCANO_QUEUE_ANNOTATION, "")
        sp = t.scheduling_policy(job)
        if sp and sp.get("queue"):
            queue = sp["queue"]
        spec = {
            "minMember": min_member,
            "queue": queue,
            "priorityClassName": calculate_priority_class_name(job),
        }
        mr = self.calculate_pg_min_resources(min_member, job)
        if mr is not None:
            spec["minResources"] = mr
        return {
            "apiVersion": f"{self.gvr.group}/{self.gvr.version}",
            "kind": "PodGroup",
            "metadata": {
                "name": t.name(job),
                "namespace": t.namespace(job),
                "ownerReferences": [t.controller_ref(job)],
            },
            "spec": spec,
        }

    def decorate_pod_template(self, tmpl: dict, job_name: str) -> None:
        tmpl.setdefault("spec", {})["schedulerName"] = self.scheduler_name
        ann = tmpl.setdefault("metadata", {}).setdefault("annotations", {})
        ann[VOLCANO_GROUP_ANNOTATION] = job_name
        # keep pod-level metadata in sync (builders copy template metadata)
        tmpl["metadata"]["annotations"] = ann


class SchedulerPluginsCtrl(PodGroupControl):
    gvr = PODGROUPS_SCHED

    def __init__(self, kube_client, scheduler_name: str = "default-scheduler",
                 priority_classes=None):
        super().__init__(kube_client, scheduler_name, priority_classes)

    def new_pod_group(self, job: dict) -> dict:
        min_member = calculate_min_available(job)
        sp = t.scheduling_policy(job)
        timeout = 0
        if sp and sp.get("scheduleTimeoutSeconds") is not None:
            timeout = int(sp["scheduleTimeoutSeconds"])
        spec = {
            "minMember": min_member,
            "scheduleTimeoutSeconds": timeout,
        }
        mr = self.calculate_pg_min_resources(min_member, job)
        if mr:
            spec["minResources"] = mr
        return {
            "apiVersion": f"{self.gvr.group}/{self.gvr.version}",
            "kind": "PodGroup",
            "metadata": {
                "name": t.name(job),
                "namespace": t.namespace(job),
                "ownerReferences": [t.controller_ref(job)],
            },
            "spec": spec,
        }

    def decorate_pod_template(self, tmpl: dict, job_name: str) -> None:
        tmpl.setdefault("spec", {})["schedulerName"] = self.scheduler_name
        labels = tmpl.setdefault("metadata", {}).setdefault("labels", {})
        labels[SCHED_PLUGINS_POD_GROUP_LABEL] = job_name
