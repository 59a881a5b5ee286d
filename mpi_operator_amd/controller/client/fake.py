"""In-memory fake KubeClient with action recording and injectable reactors —
the test double the controller unit tests drive (parity with the reference's
fake-clientset `fixture` pattern, mpi_job_controller_test.go:70-110)."""
from __future__ import annotations

import copy
import itertools
import threading
from typing import Callable, Optional

from .base import (GVR, ApiError, Conflict, KubeClient, NotFound,
                   ResourceClient, match_labels)


class Action:
    __slots__ = ("verb", "gvr", "namespace", "name", "obj")

    def __init__(self, verb, gvr, namespace, name, obj=None):
        self.verb, self.gvr, self.namespace, self.name, self.obj = \
            verb, gvr, namespace, name, obj

    def __repr__(self):
        return f"Action({self.verb} {self.gvr.resource} {self.namespace}/{self.name})"


class FakeResourceClient(ResourceClient):
    def __init__(self, parent: "FakeKubeClient", gvr: GVR):
        self.p = parent
        self.gvr = gvr

    def _key(self, ns, name):
        return (self.gvr.resource + "." + self.gvr.group, ns, name)

    def _react(self, verb, ns, name, obj):
        for r in self.p.reactors:
            handled, err = r(verb, self.gvr, ns, name, obj)
            if handled and err is not None:
                raise err
            if handled:
                return True
        return False

    def get(self, namespace, name):
        with self.p.lock:
            if self._react("get", namespace, name, None):
                return None
            o = self.p.store.get(self._key(namespace, name))
            if o is None:
                raise NotFound(f"{self.gvr.resource} {namespace}/{name}")
            return copy.deepcopy(o)

    def list(self, namespace, label_selector=None):
        with self.p.lock:
            out = []
            for (kres, ns, _), o in self.p.store.items():
                if kres == self.gvr.resource + "." + self.gvr.group and ns == namespace \
                        and match_labels(o, label_selector):
                    out.append(copy.deepcopy(o))
            return sorted(out, key=lambda o: o["metadata"]["name"])

    def create(self, namespace, obj):
        with self.p.lock:
            obj = copy.deepcopy(obj)
            name = obj["metadata"]["name"]
            self.p.actions.append(Action("create", self.gvr, namespace, name, obj))
            if self._react("create", namespace, name, obj):
                return obj
            key = self._key(namespace, name)
            if key in self.p.store:
                raise Conflict(f"{self.gvr.resource} {namespace}/{name} exists")
            obj["metadata"].setdefault("namespace", namespace)
            obj["metadata"].setdefault("uid", f"uid-{next(self.p.uid_counter)}")
            obj["metadata"]["resourceVersion"] = "1"
            obj.setdefault("apiVersion", self.gvr.api_version)
            obj.setdefault("kind", self.gvr.kind)
            self.p.store[key] = obj
            self.p.broadcast(self.gvr, "ADDED", obj)
            return copy.deepcopy(obj)

    def update(self, namespace, obj):
        with self.p.lock:
            obj = copy.deepcopy(obj)
            name = obj["metadata"]["name"]
            self.p.actions.append(Action("update", self.gvr, namespace, name, obj))
            if self._react("update", namespace, name, obj):
                return obj
            key = self._key(namespace, name)
            if key not in self.p.store:
                raise NotFound(f"{self.gvr.resource} {namespace}/{name}")
            old = self.p.store[key]
            obj["metadata"]["resourceVersion"] = str(int(old["metadata"].get("resourceVersion", "0")) + 1)
            self.p.store[key] = obj
            self.p.broadcast(self.gvr, "MODIFIED", obj)
            return copy.deepcopy(obj)

    def update_status(self, namespace, obj):
        with self.p.lock:
            obj = copy.deepcopy(obj)
            name = obj["metadata"]["name"]
            self.p.actions.append(Action("update_status", self.gvr, namespace, name, obj))
            if self._react("update_status", namespace, name, obj):
                return obj
            key = self._key(namespace, name)
            if key not in self.p.store:
                raise NotFound(f"{self.gvr.resource} {namespace}/{name}")
            cur = copy.deepcopy(self.p.store[key])
            cur["status"] = obj.get("status", {})
            cur["metadata"]["resourceVersion"] = str(int(cur["metadata"].get("resourceVersion", "0")) + 1)
            self.p.store[key] = cur
            self.p.broadcast(self.gvr, "MODIFIED", cur)
            return copy.deepcopy(cur)

    def delete(self, namespace, name):
        with self.p.lock:
            self.p.actions.append(Action("delete", self.gvr, namespace, name))
            if self._react("delete", namespace, name, None):
                return
            key = self._key(namespace, name)
            if key not in self.p.store:
                raise NotFound(f"{self.gvr.resource} {namespace}/{name}")
            obj = self.p.store.pop(key)
            self.p.broadcast(self.gvr, "DELETED", obj)

    def watch(self, namespace, resource_version=None, timeout_s: int = 60):
        """Watch-stream parity with RestResourceClient.watch: yields
        {"type": ..., "object": ...} for every mutation, starting with
        synthetic ADDED events for the current store contents (the
        list+watch informer bootstrap)."""
        q: "list" = []
        cond = threading.Condition(self.p.lock)
        with self.p.lock:
            for (kres, ns, _), o in self.p.store.items():
                if kres == self.gvr.resource + "." + self.gvr.group \
                        and (namespace is None or ns == namespace):
                    q.append({"type": "ADDED", "object": copy.deepcopy(o)})
            self.p.watchers.setdefault(self.gvr, []).append((namespace, q, cond))
        deadline = __import__("time").time() + timeout_s
        try:
            while True:
                with cond:
                    while not q:
                        remaining = deadline - __import__("time").time()
                        if remaining <= 0:
                            return
                        cond.wait(min(remaining, 0.2))
                    ev = q.pop(0)
                yield ev
        finally:
            with self.p.lock:
                try:
                    self.p.watchers[self.gvr].remove((namespace, q, cond))
                except ValueError:
                    pass


class FakeKubeClient(KubeClient):
    """reactors: callables (verb, gvr, ns, name, obj) -> (handled, exc|None)."""

    def __init__(self):
        self.store: dict = {}
        self.actions: list[Action] = []
        self.reactors: list[Callable] = []
        self.uid_counter = itertools.count(1)
        self.lock = threading.RLock()
        self._clients: dict[GVR, FakeResourceClient] = {}
        # watch plumbing: gvr -> [(namespace filter, event list, condition)]
        self.watchers: dict = {}

    def broadcast(self, gvr, ev_type: str, obj: dict) -> None:
        """Deliver a watch event to every subscriber of gvr (called with
        self.lock held by the mutating resource client)."""
        import mpi_operator_amd.controller.api.types as _t
        ns = _t.namespace(obj)
        for want_ns, q, cond in self.watchers.get(gvr, []):
            if want_ns is None or want_ns == ns:
                with cond:
                    q.append({"type": ev_type, "object": copy.deepcopy(obj)})
                    cond.notify_all()

    def resource(self, gvr: GVR) -> FakeResourceClient:
        if gvr not in self._clients:
            self._clients[gvr] = FakeResourceClient(self, gvr)
        return self._clients[gvr]

    # test helpers
    def seed(self, gvr: GVR, obj: dict) -> dict:
        return self.resource(gvr).create(obj["metadata"].get("namespace", "default"), obj)

    def actions_of(self, verb: Optional[str] = None, resource: Optional[str] = None):
        return [a for a in self.actions
                if (verb is None or a.verb == verb)
                and (resource is None or a.gvr.resource == resource)]

    def clear_actions(self):
        self.actions.clear()
