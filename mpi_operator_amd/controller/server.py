"""Operator process shell: flags, leader election (Lease), /metrics,
/healthz, watch-driven work queue, N sync workers.

Parity with reference cmd/mpi-operator/ (options.go:61-96 flags,
server.go:79-253 run/leader-election/health, main.go:29-40 metrics) built on
threads + the REST client's watch streams instead of client-go informers.
"""
from __future__ import annotations

import argparse
import logging
import os
import queue
import random
import socket
import threading
import time
from http.server import BaseHTTPRequestHandler, HTTPServer

from .api import constants as c
from .api import types as t
from .client.base import JOBS, MPIJOBS, PODS
from .client.rest import RestConfig, RestKubeClient
from .podgroup import SchedulerPluginsCtrl, VolcanoCtrl
from .reconciler import MPIJobController
from . import metrics

log = logging.getLogger("mpi-operator")


def parse_args(argv=None):
    p = argparse.ArgumentParser("mpi-operator", description="MPIJob controller (MI355X-native)")
    p.add_argument("--kubeconfig", default=os.environ.get("KUBECONFIG"))
    p.add_argument("--master", default=None, help="apiserver URL override")
    p.add_argument("--namespace",
                   default=os.environ.get(c.ENV_KUBEFLOW_NAMESPACE, ""),
                   help="namespace to watch ('' = all)")
    p.add_argument("--threadiness", type=int, default=2)
    p.add_argument("--monitoring-port", type=int, default=0,
                   help="serve prometheus /metrics on this port")
    p.add_argument("--health-port", type=int, default=8080)
    p.add_argument("--gang-scheduling", default="",
                   help="'' = off, 'volcano', or a scheduler-plugins scheduler name")
    p.add_argument("--lock-namespace", default="mpi-operator")
    p.add_argument("--kube-api-qps", type=float, default=5,
                   help="max QPS to the apiserver from this client")
    p.add_argument("--kube-api-burst", type=int, default=10,
                   help="maximum burst for the client throttle")
    p.add_argument("--controller-queue-rate-limit", type=float, default=10,
                   help="rate limit of the controller events queue")
    p.add_argument("--controller-queue-burst", type=int, default=100,
                   help="maximum burst of the controller events queue")
    p.add_argument("--cluster-domain", default="")
    p.add_argument("--leader-elect", action="store_true", default=True)
    p.add_argument("--no-leader-elect", dest="leader_elect", action="store_false")
    return p.parse_args(argv)


class HealthHandler(BaseHTTPRequestHandler):
    healthy = lambda: True  # noqa: E731 — replaced at server start

    def do_GET(self):
        if self.path == "/healthz":
            ok = type(self).healthy()
            self.send_response(200 if ok else 500)
            self.end_headers()
            self.wfile.write(b"ok" if ok else b"unhealthy")
        elif self.path == "/metrics" and metrics.HAVE_PROMETHEUS:
            from prometheus_client import generate_latest
            body = generate_latest()
            self.send_response(200)
            self.end_headers()
            self.wfile.write(body)
        else:
            self.send_response(404)
            self.end_headers()

    def log_message(self, *a):  # quiet
        pass


class LeaderElector:
    """Lease-based leader election (reference server.go:206-253;
    lease 15s / renew 5s / retry 3s)."""

    LEASE_DURATION = 15
    RENEW_PERIOD = 5
    RETRY_PERIOD = 3

    def __init__(self, client, namespace: str, name: str = "mpi-operator"):
        self.client = client.leases
        self.namespace = namespace
        self.name = name
        self.identity = f"{socket.gethostname()}_{os.getpid()}_{random.randrange(1 << 30)}"
        self.is_leader = False
        self.last_renew = 0.0

    def _lease_obj(self):
        return {
            "apiVersion": "coordination.k8s.io/v1",
            "kind": "Lease",
            "metadata": {"name": self.name, "namespace": self.namespace},
            "spec": {
                "holderIdentity": self.identity,
                "leaseDurationSeconds": self.LEASE_DURATION,
                "renewTime": t.now_iso(),
            },
        }

    def try_acquire_or_renew(self) -> bool:
        from .client.base import Conflict, NotFound
        try:
            lease = self.client.get(self.namespace, self.name)
        except NotFound:
            try:
                self.client.create(self.namespace, self._lease_obj())
                self.is_leader = True
            except Conflict:
                self.is_leader = False
            return self.is_leader
        spec = lease.get("spec", {})
        holder = spec.get("holderIdentity")
        renew = spec.get("renewTime", "1970-01-01T00:00:00Z")
        import datetime
        try:
            renew_ts = datetime.datetime.strptime(renew, "%Y-%m-%dT%H:%M:%SZ") \
                .replace(tzinfo=datetime.timezone.utc).timestamp()
        except ValueError:
            renew_ts = 0
        expired = time.time() - renew_ts > spec.get("leaseDurationSeconds", self.LEASE_DURATION)
        if holder == self.identity or expired or holder is None:
            lease["spec"] = self._lease_obj()["spec"]
            try:
                self.client.update(self.namespace, lease)
                self.is_leader = True
                self.last_renew = time.time()
            except Exception:
                self.is_leader = False
        else:
            self.is_leader = False
        return self.is_leader

    def healthy(self) -> bool:
        return (not self.is_leader) or time.time() - self.last_renew < 2 * self.LEASE_DURATION


class OperatorServer:
    """Watch-driven reconcile loop: MPIJob events and child-resource events
    (pods / launcher Jobs, one ownerRef hop) enqueue job keys; threadiness
    workers call controller.sync (reference mpi_job_controller.go:465-562,
    :1262-1312)."""

    def __init__(self, client, controller: MPIJobController, namespace: str = "",
                 threadiness: int = 2, resync_s: int = 30, rate_limiter=None):
        from .ratelimit import default_controller_limiter

        self.client = client
        self.controller = controller
        self.namespace = namespace
        self.threadiness = threadiness
        self.resync_s = resync_s
        self.queue: "queue.Queue[tuple[str, str]]" = queue.Queue()
        self._queued: set = set()
        self._lock = threading.Lock()
        self.stop = threading.Event()
        # MaxOf{per-key exponential 5ms→1000s, bucket} — a persistently
        # failing job backs off instead of hammering the apiserver
        # (reference mpi_job_controller.go:121-124)
        self.rate_limiter = rate_limiter or default_controller_limiter()

    def enqueue(self, namespace: str, name: str):
        with self._lock:
            key = (namespace, name)
            if key not in self._queued:
                self._queued.add(key)
                self.queue.put(key)

    def _owner_job_of(self, obj: dict):
        for ref in t.meta(obj).get("ownerReferences", []):
            if ref.get("kind") == "MPIJob":
                return t.namespace(obj), ref["name"]
            if ref.get("kind") == "Job" and ref.get("name", "").endswith(c.LAUNCHER_SUFFIX):
                # launcher pod → one hop through the batch Job name
                return t.namespace(obj), ref["name"][: -len(c.LAUNCHER_SUFFIX)]
        labels = t.meta(obj).get("labels", {})
        if labels.get(c.JOB_NAME_LABEL):
            return t.namespace(obj), labels[c.JOB_NAME_LABEL]
        return None

    def _handle_event(self, ev: dict, mpijob: bool):
        obj = ev.get("object", {})
        if mpijob:
            self.enqueue(t.namespace(obj), t.name(obj))
        else:
            owner = self._owner_job_of(obj)
            if owner:
                self.enqueue(*owner)
        return t.meta(obj).get("resourceVersion")

    def _watch_loop(self, gvr, mpijob: bool):
        """List once, then resume watches from the last seen resourceVersion
        (client-go ListWatch). Without the resume, every 60 s watch timeout
        re-LISTed from scratch — O(jobs) LISTs per minute per watcher.
        A 410 Gone (RV expired) falls back to a fresh list."""
        from .client.base import ApiError

        rv = None
        while not self.stop.is_set():
            try:
                rc = self.client.resource(gvr)
                if rv is None:
                    items, rv = rc.list_with_rv(self.namespace or None)
                    for obj in items:
                        self._handle_event({"object": obj}, mpijob)
                for ev in rc.watch(self.namespace or None, resource_version=rv,
                                   timeout_s=60):
                    if self.stop.is_set():
                        break
                    if ev.get("type") == "ERROR":  # in-stream 410 Gone
                        rv = None
                        break
                    rv = self._handle_event(ev, mpijob) or rv
            except ApiError as e:
                if e.code == 410:
                    rv = None  # expired RV: re-list
                else:
                    log.warning("watch %s failed: %s; retrying", gvr.resource, e)
                    time.sleep(2)
            except Exception as e:
                log.warning("watch %s failed: %s; retrying", gvr.resource, e)
                time.sleep(2)

    def _resync_loop(self):
        while not self.stop.is_set():
            try:
                for job in self.client.mpijobs.list(self.namespace or None):
                    self.enqueue(t.namespace(job), t.name(job))
            except Exception as e:
                log.warning("resync list failed: %s", e)
            self.stop.wait(self.resync_s)

    def _worker(self):
        while not self.stop.is_set():
            try:
                ns, name = self.queue.get(timeout=0.5)
            except queue.Empty:
                continue
            with self._lock:
                self._queued.discard((ns, name))
            try:
                self.controller.sync(ns, name)
                self.rate_limiter.forget((ns, name))
            except Exception as e:
                delay = self.rate_limiter.when((ns, name))
                log.warning("sync %s/%s failed: %s; requeueing in %.3fs",
                            ns, name, e, delay)
                timer = threading.Timer(delay, self.enqueue, args=(ns, name))
                timer.daemon = True
                timer.start()
            finally:
                self.queue.task_done()

    def run(self):
        threads = [
            threading.Thread(target=self._watch_loop, args=(MPIJOBS, True), daemon=True),
            threading.Thread(target=self._watch_loop, args=(PODS, False), daemon=True),
            threading.Thread(target=self._watch_loop, args=(JOBS, False), daemon=True),
            threading.Thread(target=self._resync_loop, daemon=True),
        ]
        workers = [threading.Thread(target=self._worker, daemon=True)
                   for _ in range(self.threadiness)]
        for th in threads + workers:
            th.start()
        return threads + workers


def main(argv=None):
    logging.basicConfig(level=logging.INFO,
                        format="%(asctime)s %(levelname)s %(name)s %(message)s")
    opt = parse_args(argv)
    cfg = RestConfig.from_kubeconfig(opt.kubeconfig) if opt.kubeconfig else RestConfig.auto()
    if opt.master:
        cfg.host = opt.master.rstrip("/")
    client = RestKubeClient(cfg, user_agent="mpi-operator",
                            qps=opt.kube_api_qps, burst=opt.kube_api_burst)

    if not client.crd_exists("mpijobs.kubeflow.org"):
        raise SystemExit("CRD mpijobs.kubeflow.org not found — apply manifests/ first")

    podgroup_ctrl = None
    if opt.gang_scheduling == "volcano":
        podgroup_ctrl = VolcanoCtrl(client)
    elif opt.gang_scheduling:
        podgroup_ctrl = SchedulerPluginsCtrl(client, scheduler_name=opt.gang_scheduling)

    controller = MPIJobController(client, podgroup_ctrl=podgroup_ctrl,
                                  cluster_domain=opt.cluster_domain)
    server = OperatorServer(client, controller, namespace=opt.namespace,
                            threadiness=opt.threadiness)

    elector = LeaderElector(client, opt.lock_namespace)
    HealthHandler.healthy = staticmethod(elector.healthy)
    httpd = HTTPServer(("", opt.health_port), HealthHandler)
    threading.Thread(target=httpd.serve_forever, daemon=True).start()
    if opt.monitoring_port and metrics.HAVE_PROMETHEUS:
        from prometheus_client import start_http_server
        start_http_server(opt.monitoring_port)

    if opt.leader_elect:
        log.info("waiting for leader lease as %s", elector.identity)
        while not elector.try_acquire_or_renew():
            time.sleep(LeaderElector.RETRY_PERIOD)
    metrics.is_leader.set(1)
    log.info("became leader; starting controller (threadiness=%d)", opt.threadiness)
    server.run()
    try:
        while True:
            time.sleep(LeaderElector.RENEW_PERIOD)
            if opt.leader_elect and not elector.try_acquire_or_renew():
                log.error("lost leader lease; exiting")
                raise SystemExit(1)
    except KeyboardInterrupt:
        server.stop.set()


if __name__ == "__main__":
    main()
