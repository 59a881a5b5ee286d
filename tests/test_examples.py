"""Example-workload tests: the pi C++ smoke binary under amdrun (2 ranks,
TCP reduce — the BASELINE config-1 path minus kubernetes), and the mnist
training example single-process."""
import os
import shutil
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
PI_CC = os.path.join(REPO, "examples", "v2beta1", "pi", "pi.cc")


@pytest.fixture(scope="module")
def pi_bin(tmp_path_factory):
    if shutil.which("g++") is None:
        pytest.skip("no g++")
    out = str(tmp_path_factory.mktemp("pi") / "pi")
    subprocess.check_call(["g++", "-O2", "-o", out, PI_CC])
    return out


def test_pi_two_ranks_under_amdrun(pi_bin):
    env = dict(os.environ, MASTER_PORT="29712", PYTHONPATH=REPO)
    r = subprocess.run(
        [sys.executable, "-m", "mpi_operator_amd.runtime.launcher",
         "-np", "2", "--master-port", "29712", "--", pi_bin, "200000"],
        capture_output=True, text=True, timeout=120, env=env, cwd=REPO)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "pi is approximately 3.1" in r.stdout, r.stdout


def test_mnist_example_single_process(tmp_path):
    script = os.path.join(REPO, "examples", "v2beta1", "horovod", "train_mnist.py")
    env = dict(os.environ, PYTHONPATH=REPO)
    r = subprocess.run([sys.executable, script, "--steps", "3", "--batch", "4"],
                       capture_output=True, text=True, timeout=300, env=env, cwd=str(tmp_path))
    assert r.returncode == 0, r.stdout + r.stderr
    assert "done" in r.stdout


def test_bert_example_single_process_tiny(tmp_path):
    script = os.path.join(REPO, "examples", "v2beta1", "bert", "train_bert.py")
    env = dict(os.environ, PYTHONPATH=REPO)
    r = subprocess.run([sys.executable, script, "--model", "bert_base", "--steps", "1",
                        "--warmup", "0", "--batch", "1", "--seq", "16"],
                       capture_output=True, text=True, timeout=600, env=env, cwd=str(tmp_path))
    assert r.returncode == 0, r.stdout + r.stderr
    assert "sequences/sec" in r.stdout
