"""Per-host rank agent (the orted analog): forks this host's ranks with the
rendezvous environment and reaps them, propagating the first failure."""
from __future__ import annotations

import argparse
import os
import signal
import subprocess
import sys
import time


def main(argv=None) -> int:
    ap = argparse.ArgumentParser("mpiamd-agent")
    ap.add_argument("--ranks", required=True,
                    help="comma list of global:local rank pairs, e.g. 0:0,1:1")
    ap.add_argument("--env", action="append", default=[])
    ap.add_argument("command", nargs=argparse.REMAINDER)
    args = ap.parse_args(argv)
    command = args.command
    if command and command[0] == "--":
        command = command[1:]
    base_env = dict(os.environ)
    for e in args.env:
        k, v = e.split("=", 1)
        base_env[k] = v
    ranks = [tuple(map(int, r.split(":"))) for r in args.ranks.split(",")]
    local_world = len(ranks)
    procs = []
    for rank, local_rank in ranks:
        env = dict(base_env)
        env.update({
            "RANK": str(rank),
            "LOCAL_RANK": str(local_rank),
            "LOCAL_WORLD_SIZE": str(local_world),
            # mpirun-compatible aliases (reference workloads read OMPI_*)
            "OMPI_COMM_WORLD_RANK": str(rank),
            "OMPI_COMM_WORLD_SIZE": env.get("WORLD_SIZE", "1"),
            "OMPI_COMM_WORLD_LOCAL_RANK": str(local_rank),
        })
        procs.append(subprocess.Popen(command, env=env))

    def forward(sig, _frame):
        for p in procs:
            if p.poll() is None:
                try:
                    p.send_signal(sig)
                except OSError:
                    pass

    signal.signal(signal.SIGTERM, forward)
    signal.signal(signal.SIGINT, forward)

    rc = 0
    pending = set(range(len(procs)))
    while pending:
        for i in list(pending):
            r = procs[i].poll()
            if r is None:
                continue
            pending.discard(i)
            if r != 0 and rc == 0:
                rc = r
                forward(signal.SIGTERM, None)
        time.sleep(0.2)
    return rc


if __name__ == "__main__":
    sys.exit(main())
