"""Elastic runner — the horovodrun-elastic analog over discover_hosts.sh.

Protocol parity (reference proposals/elastic-horovod.md:19-31 +
mpi_job_controller.go:1383-1407): the controller keeps a discover_hosts.sh
ConfigMap entry listing the currently-Running workers; the launcher polls it
and, on a host-set change, re-forms the job — here by terminating the ranks
and relaunching over the new host set. The workload re-enters through its
checkpoint/broadcast path (mpi_operator_amd.parallel.broadcast_parameters /
elastic_state), so training state survives the re-formation.

Env the runner exports to ranks:
  MPIAMD_ELASTIC=1, MPIAMD_RESTART_COUNT=<n> — lets the workload detect a
  re-formation and reload state from rank 0.
"""
from __future__ import annotations

import logging
import subprocess
import time

from .hostfile import HostSlots
from .launcher import Launch

log = logging.getLogger("amdrun.elastic")


def discover_hosts(script: str, slots: int) -> list[HostSlots]:
    out = subprocess.run(["/bin/sh", script], capture_output=True, text=True,
                         timeout=30)
    hosts = [line.strip() for line in out.stdout.splitlines() if line.strip()]
    return [HostSlots(h, slots) for h in hosts]


class ElasticRunner:
    def __init__(self, discover_script: str, command: list[str], master_port: int,
                 extra_env: dict, slots: int = 1, min_np: int = 1,
                 max_np: int = 1 << 30, poll_s: float = 5.0,
                 discover_fn=None, launch_cls=Launch):
        self.script = discover_script
        self.command = command
        self.master_port = master_port
        self.extra_env = dict(extra_env)
        self.slots = slots
        self.min_np, self.max_np = min_np, max_np
        self.poll_s = poll_s
        self.discover = discover_fn or (lambda: discover_hosts(self.script, self.slots))
        self.launch_cls = launch_cls
        self.restarts = 0

    def _wait_for_min(self) -> list[HostSlots]:
        while True:
            hosts = self.discover()
            if sum(h.slots for h in hosts) >= self.min_np:
                return hosts[: self.max_np]
            log.info("elastic: %d slots < min %d; waiting", sum(h.slots for h in hosts),
                     self.min_np)
            time.sleep(self.poll_s)

    def run(self) -> int:
        while True:
            hosts = self._wait_for_min()
            env = dict(self.extra_env)
            env["MPIAMD_ELASTIC"] = "1"
            env["MPIAMD_RESTART_COUNT"] = str(self.restarts)
            names = [h.host for h in hosts]
            log.info("elastic: launching over %d hosts (%s), restart #%d",
                     len(hosts), names[:4], self.restarts)
            launch = self.launch_cls(hosts, self.command, self.master_port, env).start()
            rc = self._watch(launch, names)
            if rc is not None:
                return rc
            self.restarts += 1

    def _watch(self, launch, names: list[str]):
        """Poll both the processes and the host set. Returns exit code when
        the job finishes, or None to re-form."""
        while True:
            done = all(p.poll() is not None for p in launch.procs)
            if done:
                rcs = [p.returncode for p in launch.procs]
                bad = [r for r in rcs if r != 0]
                if bad:
                    current = [h.host for h in self.discover()]
                    if current != names:
                        log.info("elastic: failure with changed host set — re-forming")
                        return None
                    return bad[0]
                return 0
            try:
                current = [h.host for h in self.discover()]
            except Exception as e:
                log.warning("elastic: discovery failed: %s", e)
                current = names
            if current != names and sum(1 for _ in current) >= self.min_np:
                log.info("elastic: host set changed %s → %s; re-forming",
                         names[:4], current[:4])
                launch.terminate()
                return None
            time.sleep(self.poll_s)
