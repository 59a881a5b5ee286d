"""Hostfile parsing — accepts both formats the controller renders
(reference mpi_job_controller.go:1347-1363):
  OpenMPI:     `host slots=N`
  Intel/MPICH: `host:N`
"""
from __future__ import annotations

from dataclasses import dataclass


@dataclass
class HostSlots:
    host: str
    slots: int


def parse_hostfile_text(text: str) -> list[HostSlots]:
    hosts: list[HostSlots] = []
    for line in text.splitlines():
        line = line.strip()
        if not line or line.startswith("#"):
            continue
        if ":" in line and " " not in line:
            host, slots = line.rsplit(":", 1)
            hosts.append(HostSlots(host, int(slots)))
            continue
        parts = line.split()
        host = parts[0]
        slots = 1
        for p in parts[1:]:
            if p.startswith("slots="):
                slots = int(p[len("slots="):])
        hosts.append(HostSlots(host, slots))
    return hosts


def parse_hostfile(path: str) -> list[HostSlots]:
    with open(path) as f:
        return parse_hostfile_text(f.read())


def total_slots(hosts: list[HostSlots]) -> int:
    return sum(h.slots for h in hosts)


def rank_assignment(hosts: list[HostSlots]) -> list[tuple[str, int, int]]:
    """[(host, global_rank, local_rank)] in hostfile order."""
    out = []
    rank = 0
    for h in hosts:
        for lr in range(h.slots):
            out.append((h.host, rank, lr))
            rank += 1
    return out
