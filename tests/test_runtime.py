"""Launcher-plane tests: amdrun local launches, rendezvous env, hostfile
parsing, elastic re-formation (local processes, CPU/gloo)."""
import os
import subprocess
import sys
import textwrap

import pytest

from mpi_operator_amd.runtime.hostfile import (HostSlots, parse_hostfile_text,
                                               rank_assignment, total_slots)
from mpi_operator_amd.runtime.elastic import ElasticRunner

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_amdrun(args, timeout=120):
    env = dict(os.environ, PYTHONPATH=REPO)
    return subprocess.run([sys.executable, "-m", "mpi_operator_amd.runtime.launcher",
                           *args], capture_output=True, text=True, timeout=timeout,
                          env=env, cwd=REPO)


def test_hostfile_formats():
    hs = parse_hostfile_text("w-0.j.ns.svc slots=8\nw-1.j.ns.svc slots=8\n")
    assert total_slots(hs) == 16
    assert rank_assignment(hs)[8] == ("w-1.j.ns.svc", 8, 0)
    hs = parse_hostfile_text("w-0.j.ns.svc:4\n# comment\n\n")
    assert hs == [HostSlots("w-0.j.ns.svc", 4)]


@pytest.mark.timeout(120)
def test_amdrun_local_ranks_env():
    code = ("import os,sys; "
            "print('R', os.environ['RANK'], os.environ['LOCAL_RANK'], "
            "os.environ['WORLD_SIZE'], os.environ['OMPI_COMM_WORLD_RANK'])")
    r = run_amdrun(["-np", "3", "--", sys.executable, "-c", code])
    assert r.returncode == 0, r.stderr
    lines = sorted(l for l in r.stdout.splitlines() if l.startswith("R "))
    assert lines == ["R 0 0 3 0", "R 1 1 3 1", "R 2 2 3 2"]


@pytest.mark.timeout(120)
def test_amdrun_nonzero_exit_propagates():
    code = "import os,sys; sys.exit(3 if os.environ['RANK']=='1' else 0)"
    r = run_amdrun(["-np", "2", "--", sys.executable, "-c", code])
    assert r.returncode == 3


@pytest.mark.timeout(180)
def test_amdrun_gloo_allreduce():
    """End-to-end: amdrun boots 2 ranks that rendezvous through our parallel
    layer (gloo here; identical code path is RCCL on MI355X)."""
    code = textwrap.dedent("""
        import torch
        from mpi_operator_amd import parallel as hvd
        hvd.init(backend="gloo")
        t = torch.full((4,), float(hvd.rank()))
        hvd.allreduce_(t)
        assert t.tolist() == [0.5]*4, t
        print("OK", hvd.rank(), hvd.size())
        hvd.shutdown()
    """)
    r = run_amdrun(["-np", "2", "--master-port", "29613", "--",
                    sys.executable, "-c", code])
    assert r.returncode == 0, r.stderr[-2000:]
    assert sorted(l for l in r.stdout.splitlines() if l.startswith("OK")) == \
        ["OK 0 2", "OK 1 2"]


@pytest.mark.timeout(180)
def test_elastic_reformation():
    """Host set changes mid-run → runner terminates and relaunches with
    MPIAMD_RESTART_COUNT bumped; completes on the second formation."""
    code = textwrap.dedent("""
        import os, sys, time
        n = int(os.environ["MPIAMD_RESTART_COUNT"])
        if n == 0:
            time.sleep(600)  # first formation only ends by re-formation
        else:
            print("DONE", os.environ["WORLD_SIZE"])
    """)
    calls = {"n": 0}

    def discover():
        calls["n"] += 1
        # host set changes by NAME after a few polls (both resolve locally)
        if calls["n"] > 2:
            return [HostSlots("127.0.0.1", 2)]
        return [HostSlots("localhost", 1)]

    runner = ElasticRunner("unused", [sys.executable, "-c", code],
                           29617, {"PYTHONPATH": REPO}, slots=1,
                           min_np=1, poll_s=0.3, discover_fn=discover)
    rc = runner.run()
    assert rc == 0
    assert runner.restarts == 1
