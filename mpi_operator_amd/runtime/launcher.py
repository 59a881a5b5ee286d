"""amdrun — the mpirun-style launcher of the MI355X-native stack.

The reference delegates process boot to OpenMPI's `mpirun` (ssh → orted →
ranks; reference mpi_job_controller.go:181-215, README.md:115-125). amdrun
re-implements that boot plane directly: it reads the controller-rendered
hostfile, SSHes one agent per host (workers run sshd, key from the MPIJob's
SSH Secret), and each agent forks the host's ranks with the rendezvous env
(RANK / WORLD_SIZE / LOCAL_RANK / MASTER_ADDR…, plus OMPI_COMM_WORLD_*
aliases). The data plane is whatever the ranks use — RCCL over xGMI for GPU
training, TCP for CPU jobs (examples/pi).

Usage:
    amdrun [-np N] [--hostfile /etc/mpi/hostfile] [--master-port P]
           [-x ENV[=VAL]]... [--elastic --discover /etc/mpi/discover_hosts.sh
           --min-np A --max-np B] -- <command> [args...]

With no hostfile: local launch with N ranks (one per GPU by default).
"""
from __future__ import annotations

import argparse
import os
import shlex
import signal
import subprocess
import sys
import time

from .hostfile import HostSlots, parse_hostfile, rank_assignment, total_slots

LOCAL_HOSTS = ("localhost", "127.0.0.1")


def _is_local(host: str) -> bool:
    import socket
    if host in LOCAL_HOSTS:
        return True
    hn = socket.gethostname()
    return host == hn or host.split(".")[0] == hn or host.split(".")[0] == os.environ.get("HOSTNAME", "\x00")


def _default_np() -> int:
    try:
        import torch
        n = torch.cuda.device_count()
        return n if n > 0 else 1
    except Exception:
        return 1


def build_env_args(pass_env: list[str]) -> dict:
    env = {}
    for e in pass_env:
        if "=" in e:
            k, v = e.split("=", 1)
            env[k] = v
        elif e in os.environ:
            env[e] = os.environ[e]
    return env


class Launch:
    def __init__(self, hosts: list[HostSlots], command: list[str], master_port: int,
                 extra_env: dict, ssh_args: list[str] | None = None):
        self.hosts = hosts
        self.command = command
        self.master_port = master_port
        self.extra_env = extra_env
        self.ssh_args = ssh_args or ["-o", "StrictHostKeyChecking=no",
                                     "-o", "UserKnownHostsFile=/dev/null",
                                     "-o", "ConnectionAttempts=10",
                                     "-o", "LogLevel=ERROR"]
        self.procs: list[subprocess.Popen] = []

    def _agent_cmd(self, host: str, ranks: list[tuple[int, int]], world: int,
                   master_addr: str) -> list[str]:
        env = dict(self.extra_env)
        env.update({
            "WORLD_SIZE": str(world),
            "MASTER_ADDR": master_addr,
            "MASTER_PORT": str(self.master_port),
            "MPIAMD_MASTER_ADDR": master_addr,
            "MPIAMD_MASTER_PORT": str(self.master_port),
            "HSA_ENABLE_IPC_MODE_LEGACY": os.environ.get("HSA_ENABLE_IPC_MODE_LEGACY", "0"),
        })
        rank_spec = ",".join(f"{r}:{lr}" for r, lr in ranks)
        agent = [sys.executable, "-m", "mpi_operator_amd.runtime.agent",
                 "--ranks", rank_spec] + \
            [f"--env={k}={v}" for k, v in env.items()] + ["--"] + self.command
        if _is_local(host):
            return agent
        remote = " ".join(shlex.quote(a) for a in agent)
        return ["ssh", *self.ssh_args, host,
                f"cd {shlex.quote(os.getcwd())} 2>/dev/null; {remote}"]

    def start(self):
        world = total_slots(self.hosts)
        master_addr = self.hosts[0].host if not _is_local(self.hosts[0].host) \
            else "127.0.0.1"
        assign = rank_assignment(self.hosts)
        by_host: dict[str, list[tuple[int, int]]] = {}
        for host, rank, lr in assign:
            by_host.setdefault(host, []).append((rank, lr))
        for host, ranks in by_host.items():
            cmd = self._agent_cmd(host, ranks, world, master_addr)
            self.procs.append(subprocess.Popen(cmd))
        return self

    def wait(self) -> int:
        """Wait for all agents; on first failure, terminate the rest."""
        rc = 0
        pending = {p.pid: p for p in self.procs}
        try:
            while pending:
                for pid, p in list(pending.items()):
                    r = p.poll()
                    if r is None:
                        continue
                    del pending[pid]
                    if r != 0 and rc == 0:
                        rc = r
                        self.terminate(exclude=pid)
                time.sleep(0.2)
        except KeyboardInterrupt:
            self.terminate()
            rc = 130
        return rc

    def terminate(self, exclude=None):
        for p in self.procs:
            if p.pid != exclude and p.poll() is None:
                try:
                    p.terminate()
                except OSError:
                    pass
        deadline = time.time() + 10
        for p in self.procs:
            if p.pid == exclude:
                continue
            try:
                p.wait(timeout=max(0.1, deadline - time.time()))
            except subprocess.TimeoutExpired:
                p.kill()


def run_once(hosts, command, master_port, extra_env) -> int:
    return Launch(hosts, command, master_port, extra_env).start().wait()


def main(argv=None) -> int:
    ap = argparse.ArgumentParser("amdrun", description=__doc__,
                                 formatter_class=argparse.RawDescriptionHelpFormatter)
    ap.add_argument("-np", type=int, default=None)
    ap.add_argument("--hostfile", default=os.environ.get("OMPI_MCA_orte_default_hostfile"))
    ap.add_argument("--master-port", type=int, default=29500)
    ap.add_argument("-x", dest="env", action="append", default=[],
                    help="pass env var (NAME or NAME=VALUE), mpirun-style")
    ap.add_argument("--elastic", action="store_true")
    ap.add_argument("--discover", default="/etc/mpi/discover_hosts.sh")
    ap.add_argument("--min-np", type=int, default=1)
    ap.add_argument("--max-np", type=int, default=1 << 30)
    ap.add_argument("--poll-s", type=float, default=5.0)
    ap.add_argument("command", nargs=argparse.REMAINDER)
    args = ap.parse_args(argv)
    command = args.command
    if command and command[0] == "--":
        command = command[1:]
    if not command:
        ap.error("no command given")
    extra_env = build_env_args(args.env)

    if args.elastic:
        from .elastic import ElasticRunner
        return ElasticRunner(args.discover, command, args.master_port, extra_env,
                             slots=args.np or 1, min_np=args.min_np,
                             max_np=args.max_np, poll_s=args.poll_s).run()

    if args.hostfile:
        hosts = parse_hostfile(args.hostfile)
        if args.np:
            # cap total ranks at np, mpirun-style
            capped, left = [], args.np
            for h in hosts:
                use = min(h.slots, left)
                if use > 0:
                    capped.append(HostSlots(h.host, use))
                    left -= use
            hosts = capped
    else:
        hosts = [HostSlots("localhost", args.np or _default_np())]
    return run_once(hosts, command, args.master_port, extra_env)


if __name__ == "__main__":
    sys.exit(main())
