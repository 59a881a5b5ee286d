"""The amdrun SSH boot plane, exercised end-to-end through the REAL
``Launch._agent_cmd`` SSH branch (reference e2e role:
test/e2e/mpi_job_test.go:87-205 boots kind + sshd + mpirun; here the
equivalent CI-runnable tier drives ssh fan-out + rank env without k8s).

Two modes:
  - real sshd when the host has one (GPU boxes / worker images);
  - otherwise an ``ssh`` PATH shim that execs the remote command via
    ``bash -lc`` — the launcher still takes the SSH branch (non-local
    host), builds the full ssh argv, quotes the remote command, and the
    agent + ranks run for real; only the transport is local.
"""
from __future__ import annotations

import json
import os
import shutil
import stat
import subprocess
import sys
import textwrap

import pytest

from mpi_operator_amd.runtime.hostfile import HostSlots
from mpi_operator_amd.runtime.launcher import Launch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _make_ssh_shim(tmp_path):
    """An ssh(1) stand-in: skips options, takes <host> <command>, runs the
    command locally. Mirrors ssh's CLI contract closely enough that the
    launcher's argv (options + host + single command string) must be
    well-formed for it to work."""
    shim = tmp_path / "ssh"
    shim.write_text(textwrap.dedent("""\
        #!/bin/bash
        # consume ssh options: -o takes a value, -p takes a value
        args=("$@")
        i=0
        while [[ $i -lt ${#args[@]} ]]; do
          case "${args[$i]}" in
            -o|-p|-i) i=$((i+2));;
            -*) i=$((i+1));;
            *) break;;
          esac
        done
        host="${args[$i]}"; i=$((i+1))
        echo "SSH_SHIM host=$host" >> "$SSH_SHIM_LOG"
        exec bash -c "${args[*]:$i}"
        """))
    shim.chmod(shim.stat().st_mode | stat.S_IEXEC)
    return shim


RANK_SCRIPT = (
    "import os, json, sys; "
    "out = os.environ['RANK_OUT']; "
    "rec = {k: os.environ.get(k) for k in ('RANK','LOCAL_RANK','WORLD_SIZE',"
    "'MASTER_ADDR','MASTER_PORT','OMPI_COMM_WORLD_RANK','OMPI_COMM_WORLD_SIZE',"
    "'OMPI_COMM_WORLD_LOCAL_RANK','PASSED_X')}; "
    "open(os.path.join(out, 'r%s.json' % os.environ['RANK']), 'w')"
    ".write(json.dumps(rec))"
)


def test_ssh_branch_fanout(tmp_path, monkeypatch):
    """2 'remote' hosts x 2 slots through the SSH branch: every rank gets
    the right RANK/WORLD_SIZE/OMPI_* env and the remote command survives
    shell quoting."""
    shim = _make_ssh_shim(tmp_path)
    out = tmp_path / "out"
    out.mkdir()
    monkeypatch.setenv("PATH", f"{tmp_path}:{os.environ['PATH']}")
    monkeypatch.setenv("SSH_SHIM_LOG", str(tmp_path / "shim.log"))
    monkeypatch.setenv("PYTHONPATH", REPO)
    monkeypatch.chdir(REPO)  # the SSH branch cd's to os.getcwd() remotely

    # fake FQDNs: not local -> the SSH branch; the shim ignores the host
    hosts = [HostSlots("w-0.job.ns.svc", 2), HostSlots("w-1.job.ns.svc", 2)]
    launch = Launch(hosts, [sys.executable, "-c", RANK_SCRIPT],
                    master_port=29871,
                    extra_env={"RANK_OUT": str(out), "PASSED_X": "quoted 'x'",
                               "PYTHONPATH": REPO})
    rc = launch.start().wait()
    assert rc == 0

    recs = {}
    for f in out.iterdir():
        recs.update({json.load(open(f))["RANK"]: json.load(open(f))})
    assert sorted(recs) == ["0", "1", "2", "3"]
    for r, rec in recs.items():
        assert rec["WORLD_SIZE"] == "4"
        assert rec["OMPI_COMM_WORLD_SIZE"] == "4"
        assert rec["OMPI_COMM_WORLD_RANK"] == r
        assert rec["LOCAL_RANK"] in ("0", "1")
        assert rec["MASTER_ADDR"] == "w-0.job.ns.svc"
        assert rec["PASSED_X"] == "quoted 'x'"  # -x quoting survives ssh
    shimlog = (tmp_path / "shim.log").read_text()
    assert "host=w-0.job.ns.svc" in shimlog
    assert "host=w-1.job.ns.svc" in shimlog


def test_ssh_branch_remote_failure_terminates_job(tmp_path, monkeypatch):
    """A failing remote rank fails the whole launch (mpirun semantics)."""
    shim = _make_ssh_shim(tmp_path)  # noqa: F841
    monkeypatch.setenv("PATH", f"{tmp_path}:{os.environ['PATH']}")
    monkeypatch.setenv("SSH_SHIM_LOG", str(tmp_path / "shim.log"))
    monkeypatch.setenv("PYTHONPATH", REPO)
    monkeypatch.chdir(REPO)
    hosts = [HostSlots("w-0.job.ns.svc", 1), HostSlots("w-1.job.ns.svc", 1)]
    launch = Launch(hosts, [sys.executable, "-c",
                            "import os,sys; sys.exit(3 if os.environ['RANK']=='1' else 0)"],
                    master_port=29872, extra_env={"PYTHONPATH": REPO})
    assert launch.start().wait() != 0


@pytest.mark.skipif(shutil.which("sshd") is None, reason="no sshd on host")
def test_real_sshd_roundtrip(tmp_path, monkeypatch):
    """Full network path when sshd exists: sshd on 127.0.0.1:23522, key
    auth, launch to 127.0.0.2 (loopback alias, non-local by name check)."""
    ssh_dir = tmp_path / "sshd"
    ssh_dir.mkdir()
    host_key = ssh_dir / "host_key"
    user_key = ssh_dir / "id_ecdsa"
    subprocess.run(["ssh-keygen", "-q", "-t", "ecdsa", "-N", "", "-f",
                    str(host_key)], check=True)
    subprocess.run(["ssh-keygen", "-q", "-t", "ecdsa", "-N", "", "-f",
                    str(user_key)], check=True)
    auth = ssh_dir / "authorized_keys"
    auth.write_text(user_key.with_suffix(".pub").read_text())
    auth.chmod(0o600)
    cfg = ssh_dir / "sshd_config"
    cfg.write_text(textwrap.dedent(f"""\
        Port 23522
        ListenAddress 127.0.0.1
        HostKey {host_key}
        AuthorizedKeysFile {auth}
        PidFile {ssh_dir}/sshd.pid
        StrictModes no
        PasswordAuthentication no
        UsePAM no
        """))
    sshd = subprocess.Popen([shutil.which("sshd"), "-D", "-f", str(cfg)])
    try:
        import time
        for _ in range(50):
            r = subprocess.run(["ssh", "-p", "23522", "-i", str(user_key),
                                "-o", "StrictHostKeyChecking=no",
                                "-o", "UserKnownHostsFile=/dev/null",
                                "-o", "LogLevel=ERROR", "127.0.0.2", "true"])
            if r.returncode == 0:
                break
            time.sleep(0.2)
        else:
            pytest.skip("sshd did not come up / loopback auth refused")
        out = tmp_path / "out"
        out.mkdir()
        monkeypatch.chdir(REPO)
        hosts = [HostSlots("127.0.0.2", 2)]
        launch = Launch(hosts, [sys.executable, "-c", RANK_SCRIPT],
                        master_port=29873,
                        extra_env={"RANK_OUT": str(out), "PASSED_X": "y",
                                   "PYTHONPATH": REPO},
                        ssh_args=["-p", "23522", "-i", str(user_key),
                                  "-o", "StrictHostKeyChecking=no",
                                  "-o", "UserKnownHostsFile=/dev/null",
                                  "-o", "LogLevel=ERROR"])
        assert launch.start().wait() == 0
        ranks = sorted(json.load(open(f))["RANK"] for f in out.iterdir())
        assert ranks == ["0", "1"]
    finally:
        sshd.terminate()
        sshd.wait(timeout=10)
