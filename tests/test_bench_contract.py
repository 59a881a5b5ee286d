"""The driver contract on bench.py: runs on CPU (tiny shapes), prints ONE
JSON line with the required fields, and supports the --impl torch A/B."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_bench(*args):
    env = dict(os.environ, PYTHONPATH=REPO)
    r = subprocess.run([sys.executable, os.path.join(REPO, "bench.py"),
                        "--steps", "2", "--warmup", "1", *args],
                       capture_output=True, text=True, timeout=600, env=env)
    assert r.returncode == 0, r.stdout + r.stderr
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout
    return json.loads(lines[0])


def test_bench_json_contract():
    out = run_bench()
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in out, key
    assert out["metric"] == "images/sec"
    assert out["n_gpus"] == 1
    assert out["steps"] == 2
    assert out["scaling"] == "weak"
    assert out["data"] == "synthetic"
    assert out["config"]["model"] == "resnet101"
    assert out["value"] > 0


def test_bench_torch_impl_ab_path():
    out = run_bench("--impl", "torch", "--model", "resnet50")
    assert out["config"]["impl"] == "torch"
    assert out["value"] > 0


def test_bench_under_torchrun_two_ranks():
    """The driver's exact N>1 launch shape: torchrun, one JSON from rank 0,
    whole-job aggregate value, allreduce over gloo."""
    env = dict(os.environ, PYTHONPATH=REPO)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29641", os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=600, env=env)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout  # rank 0 only
    out = json.loads(lines[0])
    assert out["n_gpus"] == 2
    assert out["config"]["parallelism"] == "dp2"


def test_bench_under_torchrun_eight_ranks():
    """the 8-rank shape of the driver's scaling run,
    exercised on CPU/gloo so the first 8-GPU execution is not the first
    8-rank execution of the bucket-overlap + MAX-over-ranks path."""
    env = dict(os.environ, PYTHONPATH=REPO)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
         "--master-port", "29642", os.path.join(REPO, "bench.py"),
         "--gpus", "8", "--steps", "1", "--warmup", "0",
         "--model", "resnet50"],
        capture_output=True, text=True, timeout=900, env=env)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout
    out = json.loads(lines[0])
    assert out["n_gpus"] == 8
    assert out["config"]["parallelism"] == "dp8"
