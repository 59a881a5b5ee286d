"""MPIJobController — the reconcile engine.

Faithful re-implementation of the reference sync algorithm
(pkg/controller/mpi_job_controller.go:567-741 `syncHandler`, with the
get-or-create/ownership/lifecycle helpers :743-1092 and the status manager
:1094-1233), operating on dict-form objects through the KubeClient interface
(fake in tests, REST in production)."""
from __future__ import annotations

import copy
import logging

from .api import constants as c
from .api import defaults, types as t, validation
from . import builders, metrics, status as st

log = logging.getLogger("mpi-operator")


class EventRecorder:
    """Creates corev1 Events attributed to the MPIJob (reference uses
    record.EventRecorder; asserted by tests the same way)."""

    def __init__(self, client):
        self.client = client
        self._seq = 0

    def event(self, job: dict, etype: str, reason: str, message: str) -> None:
        self._seq += 1
        ev = {
            "apiVersion": "v1",
            "kind": "Event",
            "metadata": {
                "name": f"{t.name(job)}.{self._seq:x}",
                "namespace": t.namespace(job),
            },
            "involvedObject": {
                "apiVersion": c.API_GROUP_VERSION,
                "kind": c.KIND,
                "name": t.name(job),
                "namespace": t.namespace(job),
                "uid": t.uid(job),
            },
            "reason": reason,
            "message": message,
            "type": etype,
            "source": {"component": c.CONTROLLER_AGENT_NAME},
            "firstTimestamp": t.now_iso(),
            "lastTimestamp": t.now_iso(),
            "count": 1,
        }
        try:
            self.client.events.create(t.namespace(job), ev)
        except Exception:  # events are best-effort
            log.exception("failed to record event")


def truncate_message(msg: str) -> str:
    if len(msg) <= c.EVENT_MESSAGE_LIMIT:
        return msg
    return msg[: c.EVENT_MESSAGE_LIMIT - 3] + "..."


def _job_condition(job_obj: dict, cond_type: str) -> dict | None:
    for cond in t.deep_get(job_obj, "status", "conditions", default=[]):
        if cond.get("type") == cond_type:
            return cond
    return None


def is_batch_job_finished(job_obj: dict) -> bool:
    return is_batch_job_succeeded(job_obj) or is_batch_job_failed(job_obj)


def is_batch_job_succeeded(job_obj: dict) -> bool:
    cond = _job_condition(job_obj, "Complete")
    return cond is not None and cond.get("status") == "True"


def is_batch_job_failed(job_obj: dict) -> bool:
    cond = _job_condition(job_obj, "Failed")
    return cond is not None and cond.get("status") == "True"


def is_batch_job_suspended(job_obj: dict) -> bool:
    return bool(t.deep_get(job_obj, "spec", "suspend", default=False))


def pod_phase(p: dict) -> str:
    return t.deep_get(p, "status", "phase", default="")


def is_pod_ready(p: dict) -> bool:
    for cond in t.deep_get(p, "status", "conditions", default=[]):
        if cond.get("type") == "Ready" and cond.get("status") == "True":
            return True
    return False


class MPIJobController:
    def __init__(self, client, podgroup_ctrl=None, cluster_domain: str = "",
                 keygen=builders.generate_ssh_keypair, now=None):
        self.client = client
        self.podgroup_ctrl = podgroup_ctrl
        self.cluster_domain = cluster_domain
        self.keygen = keygen
        self.recorder = EventRecorder(client)
        self.now = now or t.now_iso  # injectable clock (tests)

    # ------------------------------------------------------------------ sync
    def sync(self, namespace: str, name: str) -> None:
        """One reconcile pass for namespace/name (syncHandler :567-741)."""
        from .client.base import NotFound

        try:
            shared = self.client.mpijobs.get(namespace, name)
        except NotFound:
            log.debug("MPIJob deleted: %s/%s", namespace, name)
            return
        job = copy.deepcopy(shared)
        defaults.set_defaults_mpijob(job)

        if t.managed_by(job) != c.KUBEFLOW_JOB_CONTROLLER:
            log.info("Skipping MPIJob managed by %s", t.managed_by(job))
            return
        if t.meta(job).get("deletionTimestamp"):
            return
        errs = validation.validate_mpijob(job)
        if errs:
            msg = truncate_message(f"Found validation errors: {'; '.join(errs)}")
            self.recorder.event(job, "Warning", c.VALIDATION_ERROR, msg)
            return  # do not requeue

        if not t.status(job).get("conditions"):
            msg = f"MPIJob {namespace}/{name} is created."
            st.update_job_conditions(job, c.JOB_CREATED, "True", st.REASON_CREATED,
                                     msg, self.now())
            self.recorder.event(job, "Normal", "MPIJobCreated", msg)
            metrics.jobs_created_total.inc()

        # finished && CompletionTime → cleanup then stop
        if st.is_finished(t.status(job)) and t.status(job).get("completionTime"):
            if t.clean_pod_policy(job) in (c.CLEAN_POD_POLICY_ALL, c.CLEAN_POD_POLICY_RUNNING):
                self._clean_up_worker_pods(job)
                self._update_status_subresource(job)
            return

        if t.status(job).get("startTime") is None and not t.is_suspended(job):
            t.status(job)["startTime"] = self.now()

        launcher = self._get_launcher_job(job)

        workers: list = []
        done = launcher is not None and is_batch_job_finished(launcher)
        if not done:
            self._get_or_create_service(job, builders.new_job_service(job))
            self._get_or_create_config_map(job)
            self._get_or_create_ssh_auth_secret(job)
            if not t.is_suspended(job):
                if self.podgroup_ctrl is not None:
                    self._get_or_create_pod_group(job)
                workers = self._get_or_create_workers(job)
            if launcher is None:
                # WaitForWorkersReady gates on the DESIRED replica count:
                # comparing against the created list would let a suspended
                # job (workers == []) create its launcher early
                if t.launcher_creation_policy(job) == c.LAUNCHER_CREATION_AT_STARTUP \
                        or self._count_ready(workers) == t.worker_replicas(job):
                    try:
                        launcher = self.client.jobs.create(
                            namespace,
                            builders.new_launcher_job(
                                job, self.podgroup_ctrl,
                                lambda r, m: self.recorder.event(job, "Warning", r, m)))
                    except Exception as e:
                        self.recorder.event(job, "Warning", st.REASON_FAILED,
                                            f"launcher pod created failed: {e}")
                        raise
                else:
                    log.debug("Waiting for workers %s/%s to start", namespace, name)

        if launcher is not None:
            if not t.is_suspended(job) and is_batch_job_suspended(launcher):
                # resume: clear StartTime via status subresource first (Job
                # template is immutable once StartTime set), sync KEP-2926
                # mutable scheduling directives, unsuspend (:690-724)
                lcopy = copy.deepcopy(launcher)
                if t.deep_get(lcopy, "status", "startTime") is not None:
                    lcopy["status"]["startTime"] = None
                    lcopy = self.client.jobs.update_status(namespace, lcopy)
                desired = builders.new_launcher_pod_template(job, self.podgroup_ctrl)
                self._sync_scheduling_directives(lcopy, desired)
                lcopy["spec"]["suspend"] = False
                launcher = self.client.jobs.update(namespace, lcopy)
            elif t.is_suspended(job) and not is_batch_job_suspended(launcher):
                lcopy = copy.deepcopy(launcher)
                lcopy["spec"]["suspend"] = True
                launcher = self.client.jobs.update(namespace, lcopy)

        if t.is_suspended(job):
            self._clean_up_worker_pods(job)

        self._update_mpijob_status(job, launcher, workers)

    # ----------------------------------------------------------- helpers
    def _ownership_guard(self, obj: dict, job: dict, kind: str):
        if not t.controlled_by(obj, job):
            msg = f'Resource "{t.name(obj)}" of Kind "{kind}" already exists and is not managed by MPIJob'
            self.recorder.event(job, "Warning", c.ERR_RESOURCE_EXISTS, msg)
            raise RuntimeError(msg)

    def _get_launcher_job(self, job: dict):
        from .client.base import NotFound
        try:
            launcher = self.client.jobs.get(t.namespace(job), t.launcher_name(job))
        except NotFound:
            return None
        self._ownership_guard(launcher, job, "Job")
        return launcher

    def _get_or_create_service(self, job: dict, new_svc: dict):
        from .client.base import NotFound
        ns = t.namespace(job)
        try:
            svc = self.client.services.get(ns, t.name(new_svc))
        except NotFound:
            return self.client.services.create(ns, new_svc)
        self._ownership_guard(svc, job, "Service")
        if svc["spec"].get("selector") != new_svc["spec"]["selector"] or \
                svc["spec"].get("publishNotReadyAddresses", False) != \
                new_svc["spec"]["publishNotReadyAddresses"]:
            svc = copy.deepcopy(svc)
            svc["spec"]["selector"] = new_svc["spec"]["selector"]
            svc["spec"]["publishNotReadyAddresses"] = new_svc["spec"]["publishNotReadyAddresses"]
            return self.client.services.update(ns, svc)
        return svc

    def _running_worker_pods(self, job: dict) -> list:
        pods = self.client.pods.list(t.namespace(job),
                                     builders.worker_selector(t.name(job)))
        return [p for p in pods if pod_phase(p) == "Running"]

    def _get_or_create_config_map(self, job: dict):
        from .client.base import NotFound
        ns = t.namespace(job)
        new_cm = builders.new_config_map(job, t.worker_replicas(job), self.cluster_domain)
        builders.update_discover_hosts(new_cm, job, self._running_worker_pods(job),
                                       self.cluster_domain)
        try:
            cm = self.client.configmaps.get(ns, t.config_name(job))
        except NotFound:
            return self.client.configmaps.create(ns, new_cm)
        self._ownership_guard(cm, job, "ConfigMap")
        if cm.get("data") != new_cm["data"]:
            cm = copy.deepcopy(cm)
            cm["data"] = new_cm["data"]
            return self.client.configmaps.update(ns, cm)
        return cm

    def _get_or_create_ssh_auth_secret(self, job: dict):
        from .client.base import NotFound
        ns = t.namespace(job)
        try:
            secret = self.client.secrets.get(ns, t.ssh_secret_name(job))
        except NotFound:
            return self.client.secrets.create(
                ns, builders.new_ssh_auth_secret(job, self.keygen))
        self._ownership_guard(secret, job, "Secret")
        # keys (not values) must match; regenerate only if the shape changed
        want = sorted(["ssh-privatekey", c.SSH_PUBLIC_KEY])
        if sorted((secret.get("data") or {}).keys()) != want:
            new_secret = builders.new_ssh_auth_secret(job, self.keygen)
            secret = copy.deepcopy(secret)
            secret["data"] = new_secret["data"]
            return self.client.secrets.update(ns, secret)
        return secret

    def _get_or_create_pod_group(self, job: dict):
        from .client.base import NotFound
        ns = t.namespace(job)
        new_pg = self.podgroup_ctrl.new_pod_group(job)
        try:
            pg = self.podgroup_ctrl.client.get(ns, t.name(job))
        except NotFound:
            return self.podgroup_ctrl.client.create(ns, new_pg)
        self._ownership_guard(pg, job, "PodGroup")
        if not self.podgroup_ctrl.pg_specs_equal(pg, new_pg):
            pg = copy.deepcopy(pg)
            pg["spec"] = new_pg["spec"]
            return self.podgroup_ctrl.client.update(ns, pg)
        return pg

    def _delete_pod_group(self, job: dict):
        from .client.base import NotFound
        try:
            pg = self.podgroup_ctrl.client.get(t.namespace(job), t.name(job))
        except NotFound:
            return
        self._ownership_guard(pg, job, "PodGroup")
        self.podgroup_ctrl.client.delete(t.namespace(job), t.name(job))

    def _get_or_create_workers(self, job: dict) -> list:
        from .client.base import NotFound
        ns = t.namespace(job)
        worker_spec = t.worker_spec(job)
        if worker_spec is None:
            return []
        replicas = int(worker_spec.get("replicas", 0))
        # scale-down: delete pods with index >= replicas (:998-1014, elastic)
        pods = self.client.pods.list(ns, builders.worker_selector(t.name(job)))
        if len(pods) > replicas:
            for p in pods:
                idx_str = t.meta(p).get("labels", {}).get(c.REPLICA_INDEX_LABEL)
                if idx_str is None:
                    continue
                try:
                    idx = int(idx_str)
                except ValueError:
                    continue
                if t.run_launcher_as_worker(job):
                    idx -= 1  # index label is padded by one
                if idx >= replicas:
                    self.client.pods.delete(ns, t.name(p))
        workers = []
        for i in range(replicas):
            try:
                pod = self.client.pods.get(ns, t.worker_name(job, i))
            except NotFound:
                try:
                    pod = self.client.pods.create(
                        ns, builders.new_worker(job, i, self.podgroup_ctrl))
                except Exception as e:
                    self.recorder.event(job, "Warning", st.REASON_FAILED,
                                        f"worker pod created failed: {e}")
                    raise
            self._ownership_guard(pod, job, "Pod")
            workers.append(pod)
        return workers

    def _delete_worker_pods(self, job: dict):
        from .client.base import NotFound
        ns = t.namespace(job)
        worker_spec = t.worker_spec(job)
        if worker_spec is None:
            return
        policy = t.clean_pod_policy(job)
        for i in range(int(worker_spec.get("replicas", 0))):
            name = t.worker_name(job, i)
            try:
                pod = self.client.pods.get(ns, name)
            except NotFound:
                continue
            self._ownership_guard(pod, job, "Pod")
            # Running policy keeps pods that are neither running nor pending
            if policy == c.CLEAN_POD_POLICY_RUNNING and \
                    pod_phase(pod) not in ("Running", "Pending"):
                continue
            try:
                self.client.pods.delete(ns, name)
            except NotFound:
                pass

    def _clean_up_worker_pods(self, job: dict):
        self._delete_worker_pods(job)
        st.initialize_replica_statuses(job, c.MPI_REPLICA_TYPE_WORKER)
        if self.podgroup_ctrl is not None:
            self._delete_pod_group(job)
        t.status(job)["replicaStatuses"][c.MPI_REPLICA_TYPE_WORKER]["active"] = 0

    @staticmethod
    def _count_ready(workers: list) -> int:
        return sum(1 for w in workers if is_pod_ready(w))

    @staticmethod
    def _sync_scheduling_directives(launcher: dict, desired_tmpl: dict):
        """KEP-2926 mutable scheduling directives on resume (:1685-1692)."""
        tmpl = launcher["spec"].setdefault("template", {})
        dspec = desired_tmpl.get("spec", {})
        tspec = tmpl.setdefault("spec", {})
        for f in ("nodeSelector", "tolerations", "schedulingGates", "affinity"):
            if f in dspec:
                tspec[f] = dspec[f]
            else:
                tspec.pop(f, None)
        for f in ("labels", "annotations"):
            v = desired_tmpl.get("metadata", {}).get(f)
            if v is not None:
                tmpl.setdefault("metadata", {})[f] = v

    # ----------------------------------------------------------- status
    def _launcher_pods(self, launcher: dict) -> list:
        sel = t.deep_get(launcher, "spec", "selector", "matchLabels")
        pods = self.client.pods.list(t.namespace(launcher), sel)
        return [p for p in pods if sel or t.controlled_by(p, launcher)]

    def _update_mpijob_status(self, job: dict, launcher, workers: list):
        old_status = copy.deepcopy(t.status(job))
        if t.is_suspended(job):
            if st.update_job_conditions(job, c.JOB_SUSPENDED, "True",
                                        st.REASON_SUSPENDED, "MPIJob suspended",
                                        self.now()):
                self.recorder.event(job, "Normal", "MPIJobSuspended", "MPIJob suspended")
        elif t.get_condition(t.status(job), c.JOB_SUSPENDED) is not None:
            if st.update_job_conditions(job, c.JOB_SUSPENDED, "False",
                                        st.REASON_RESUMED, "MPIJob resumed",
                                        self.now()):
                self.recorder.event(job, "Normal", "MPIJobResumed", "MPIJob resumed")
                t.status(job)["startTime"] = self.now()

        ns, name = t.namespace(job), t.name(job)
        launcher_pods_cnt = 0
        if launcher is not None:
            launcher_pods = self._launcher_pods(launcher)
            launcher_pods_cnt = sum(1 for p in launcher_pods if pod_phase(p) == "Running")
            st.initialize_replica_statuses(job, c.MPI_REPLICA_TYPE_LAUNCHER)
            lstatus = t.status(job)["replicaStatuses"][c.MPI_REPLICA_TYPE_LAUNCHER]
            failed = t.deep_get(launcher, "status", "failed", default=0)
            if failed:
                lstatus["failed"] = failed
            if is_batch_job_succeeded(launcher):
                lstatus["succeeded"] = 1
                msg = f"MPIJob {ns}/{name} successfully completed."
                self.recorder.event(job, "Normal", st.REASON_SUCCEEDED, msg)
                if not t.status(job).get("completionTime"):
                    t.status(job)["completionTime"] = \
                        t.deep_get(launcher, "status", "completionTime") or self.now()
                st.update_job_conditions(job, c.JOB_SUCCEEDED, "True",
                                         st.REASON_SUCCEEDED, msg, self.now())
                metrics.jobs_successful_total.inc()
            elif is_batch_job_failed(launcher):
                self._update_failed_status(job, launcher, launcher_pods)
            else:
                lstatus["active"] = launcher_pods_cnt
            metrics.job_info.labels(t.name(launcher), ns).set(1)

        running = evict = 0
        st.initialize_replica_statuses(job, c.MPI_REPLICA_TYPE_WORKER)
        wstatus = t.status(job)["replicaStatuses"][c.MPI_REPLICA_TYPE_WORKER]
        for w in workers:
            phase = pod_phase(w)
            if phase == "Failed":
                wstatus["failed"] = wstatus.get("failed", 0) + 1
                if t.deep_get(w, "status", "reason") == "Evicted":
                    evict += 1
            elif phase == "Succeeded":
                wstatus["succeeded"] = wstatus.get("succeeded", 0) + 1
            elif phase == "Running":
                running += 1
                wstatus["active"] = wstatus.get("active", 0) + 1
        if evict > 0:
            msg = f"{evict}/{len(workers)} workers are evicted"
            st.update_job_conditions(job, c.JOB_FAILED, "True", st.REASON_EVICTED,
                                     msg, self.now())
            self.recorder.event(job, "Warning", st.REASON_EVICTED, msg)

        if t.is_suspended(job):
            st.update_job_conditions(job, c.JOB_RUNNING, "False", st.REASON_SUSPENDED,
                                     f"MPIJob {ns}/{name} is suspended.", self.now())
        elif st.is_finished(t.status(job)):
            if t.get_condition(t.status(job), c.JOB_RUNNING) is None:
                when = t.status(job).get("completionTime") or self.now()
                t.status(job).setdefault("conditions", []).append({
                    "type": c.JOB_RUNNING, "status": "False",
                    "reason": st.REASON_RUNNING,
                    "message": f"MPIJob {ns}/{name} is finished but Running "
                               f"condition was never set.",
                    "lastUpdateTime": when, "lastTransitionTime": when,
                })
        elif launcher is not None and launcher_pods_cnt >= 1 and running == len(workers):
            msg = f"MPIJob {ns}/{name} is running."
            if st.update_job_conditions(job, c.JOB_RUNNING, "True",
                                        st.REASON_RUNNING, msg, self.now()):
                self.recorder.event(job, "Normal", "MPIJobRunning",
                                    f"MPIJob {ns}/{name} is running")

        if old_status != t.status(job):
            self._update_status_subresource(job)

    def _update_failed_status(self, job: dict, launcher: dict, launcher_pods: list):
        cond = _job_condition(launcher, "Failed") or {}
        reason = cond.get("reason") or st.REASON_FAILED
        msg = cond.get("message") or \
            f"MPIJob {t.namespace(job)}/{t.name(job)} has failed"
        if reason == "BackoffLimitExceeded":
            failed_pods = [p for p in launcher_pods if pod_phase(p) == "Failed"]
            failed_pods.sort(key=lambda p: t.meta(p).get("creationTimestamp", ""))
            if failed_pods:
                last = failed_pods[-1]
                reason += "/" + t.deep_get(last, "status", "reason", default="")
                msg += ": " + t.deep_get(last, "status", "message", default="")
                msg = truncate_message(msg)
        self.recorder.event(job, "Warning", reason, msg)
        if not t.status(job).get("completionTime"):
            t.status(job)["completionTime"] = self.now()
        st.update_job_conditions(job, c.JOB_FAILED, "True", reason, msg, self.now())
        metrics.jobs_failed_total.inc()

    def _update_status_subresource(self, job: dict):
        self.client.mpijobs.update_status(t.namespace(job), job)
