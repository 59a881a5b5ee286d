"""JobStatus condition machinery — parity with reference
pkg/controller/mpi_job_controller_status.go:41-144."""
from __future__ import annotations

from .api import constants as c
from .api import types as t

REASON_CREATED = "MPIJobCreated"
REASON_SUCCEEDED = "MPIJobSucceeded"
REASON_RUNNING = "MPIJobRunning"
REASON_SUSPENDED = "MPIJobSuspended"
REASON_RESUMED = "MPIJobResumed"
REASON_FAILED = "MPIJobFailed"
REASON_EVICTED = "MPIJobEvicted"


def initialize_replica_statuses(job: dict, rtype: str) -> None:
    st = t.status(job)
    st.setdefault("replicaStatuses", {})[rtype] = {}


def update_job_conditions(job: dict, cond_type: str, status: str, reason: str,
                          message: str, now=None) -> bool:
    now = now or t.now_iso()
    cond = {
        "type": cond_type,
        "status": status,
        "lastUpdateTime": now,
        "lastTransitionTime": now,
        "reason": reason,
        "message": message,
    }
    return _set_condition(t.status(job), cond)


def _set_condition(st: dict, cond: dict) -> bool:
    cur = t.get_condition(st, cond["type"])
    if cur is not None and cur.get("status") == cond["status"] \
            and cur.get("reason") == cond["reason"]:
        return False
    if cur is not None and cur.get("status") == cond["status"]:
        cond["lastTransitionTime"] = cur.get("lastTransitionTime")
    st["conditions"] = _filter_out(st.get("conditions", []), cond["type"]) + [cond]
    return True


def _filter_out(conditions: list, cond_type: str) -> list:
    out = []
    for cond in conditions:
        # Running and Restarting are mutually exclusive
        if cond_type == c.JOB_RESTARTING and cond.get("type") == c.JOB_RUNNING:
            continue
        if cond_type == c.JOB_RUNNING and cond.get("type") == c.JOB_RESTARTING:
            continue
        if cond.get("type") == cond_type:
            continue
        cond = dict(cond)
        # terminal conditions force Running/Failed to False
        if cond_type in (c.JOB_FAILED, c.JOB_SUCCEEDED) and \
                cond.get("type") in (c.JOB_RUNNING, c.JOB_FAILED):
            cond["status"] = "False"
        out.append(cond)
    return out


def is_finished(st: dict) -> bool:
    return is_succeeded(st) or is_failed(st)


def is_succeeded(st: dict) -> bool:
    return t.has_condition_true(st, c.JOB_SUCCEEDED)


def is_failed(st: dict) -> bool:
    return t.has_condition_true(st, c.JOB_FAILED)
