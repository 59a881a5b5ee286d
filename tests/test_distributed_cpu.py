"""Multi-process (world_size=2, gloo) tests of the Horovod-equivalent layer:
rendezvous, broadcast, DistributedOptimizer bucketized allreduce. These run
on CPU here and exercise the same code path RCCL uses on an MI355X node."""
import os

import pytest
import torch
import torch.multiprocessing as mp

PORT = 29611


def _dist_env(rank, world, port):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)


def _worker_allreduce(rank, world, port, q):
    try:
        _dist_env(rank, world, port)
        from mpi_operator_amd import parallel as hvd
        from mpi_operator_amd.parallel import DistributedOptimizer

        hvd.init(backend="gloo")
        assert hvd.rank() == rank and hvd.size() == world

        torch.manual_seed(42)  # same init on both ranks
        m = torch.nn.Sequential(torch.nn.Linear(32, 64), torch.nn.ReLU(),
                                torch.nn.Linear(64, 8))
        hvd.broadcast_parameters(m, root_rank=0)
        opt = torch.optim.SGD(m.parameters(), lr=0.1)
        dopt = DistributedOptimizer(opt, bucket_bytes=4096)  # force >1 bucket

        torch.manual_seed(100 + rank)  # different data per rank
        x = torch.randn(4, 32)
        y = m(x).pow(2).mean()
        dopt.zero_grad()
        y.backward()
        dopt.synchronize()

        # grads must now equal the average of both ranks' local grads:
        # recompute local grads on a twin model for comparison
        m2 = torch.nn.Sequential(torch.nn.Linear(32, 64), torch.nn.ReLU(),
                                 torch.nn.Linear(64, 8))
        m2.load_state_dict(m.state_dict())
        grads_local = torch.autograd.grad(m2(x).pow(2).mean(), m2.parameters())
        import torch.distributed as dist
        for p, gl in zip(m.parameters(), grads_local):
            avg = gl.clone()
            dist.all_reduce(avg)
            avg /= world
            assert torch.allclose(p.grad, avg, atol=1e-6), (p.grad - avg).abs().max()

        # synchronize() must be idempotent: step() re-enters it (the Horovod
        # synchronize-then-step pattern) and must NOT average a second time
        snap = [p.grad.clone() for p in m.parameters()]
        dopt.synchronize()
        for p, s in zip(m.parameters(), snap):
            assert torch.equal(p.grad, s), "double-averaged grads"

        dopt.step()
        # params identical across ranks after the averaged step
        flat = torch.cat([p.data.flatten() for p in m.parameters()])
        mine = flat.clone()
        dist.broadcast(flat, src=0)
        assert torch.allclose(mine, flat, atol=1e-6)
        hvd.shutdown()
        q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        import traceback
        q.put((rank, f"FAIL {e}\n{traceback.format_exc()}"))


def _worker_bcast_object(rank, world, port, q):
    try:
        _dist_env(rank, world, port)
        from mpi_operator_amd import parallel as hvd

        hvd.init(backend="gloo")
        got = hvd.broadcast_object({"epoch": 7} if rank == 0 else None, root_rank=0)
        assert got == {"epoch": 7}
        t = torch.full((4,), float(rank))
        hvd.allreduce_(t)
        assert torch.allclose(t, torch.full((4,), sum(range(world)) / world))
        hvd.shutdown()
        q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        import traceback
        q.put((rank, f"FAIL {e}\n{traceback.format_exc()}"))


def _run_workers(fn, world=2, port=PORT):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=fn, args=(r, world, port, q)) for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get() for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"


@pytest.mark.timeout(180)
def test_distributed_optimizer_allreduce():
    _run_workers(_worker_allreduce, port=PORT)


@pytest.mark.timeout(180)
def test_broadcast_and_allreduce():
    _run_workers(_worker_bcast_object, port=PORT + 1)

def _worker_checkpoint(rank, world, port, q):
    try:
        _dist_env(rank, world, port)
        import tempfile
        from mpi_operator_amd import parallel as hvd
        from mpi_operator_amd.parallel import checkpoint as ckpt

        hvd.init(backend="gloo")
        tmpdir = hvd.broadcast_object(
            tempfile.mkdtemp() if rank == 0 else None)
        path = os.path.join(tmpdir, "ck.pt")

        torch.manual_seed(5 + rank)  # deliberately DIFFERENT init per rank
        m = torch.nn.Linear(16, 4)
        opt = torch.optim.SGD(m.parameters(), lr=0.1, momentum=0.9)
        m(torch.randn(2, 16)).sum().backward()
        opt.step()

        # rank-0 state becomes the checkpoint
        ckpt.save_checkpoint(path, m, opt, epoch=3, step=77)
        assert os.path.exists(path)  # barrier in save => visible on all ranks

        # fresh model with different weights on every rank; restore must
        # converge everyone onto rank-0's saved state
        torch.manual_seed(90 + rank)
        m2 = torch.nn.Linear(16, 4)
        opt2 = torch.optim.SGD(m2.parameters(), lr=0.1, momentum=0.9)
        extra = ckpt.load_checkpoint(path, m2, opt2)
        assert extra == {"epoch": 3, "step": 77}

        import torch.distributed as dist
        flat = torch.cat([p.data.flatten() for p in m2.parameters()])
        ref = flat.clone()
        dist.broadcast(ref, src=0)
        assert torch.allclose(flat, ref, atol=0), "params differ across ranks"
        # momentum buffers restored too
        n_buf = sum(1 for s in opt2.state.values() if "momentum_buffer" in s)
        assert n_buf == len(list(m2.parameters()))

        # missing file => fresh start, returns {}
        assert ckpt.load_checkpoint(os.path.join(tmpdir, "none.pt"), m2) == {}
        hvd.shutdown()
        if rank == 0:
            import shutil
            shutil.rmtree(tmpdir, ignore_errors=True)
        q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        import traceback
        q.put((rank, f"FAIL {e}\n{traceback.format_exc()}"))


@pytest.mark.timeout(180)
def test_checkpoint_rank0_save_broadcast_restore():
    _run_workers(_worker_checkpoint, port=PORT + 2)


def test_checkpoint_single_process(tmp_path):
    """World-1 path (no process group): plain save/load roundtrip."""
    from mpi_operator_amd.parallel import checkpoint as ckpt
    m = torch.nn.Linear(8, 2)
    opt = torch.optim.SGD(m.parameters(), lr=0.01)
    p = str(tmp_path / "ck.pt")
    ckpt.save_checkpoint(p, m, opt, epoch=1)
    m2 = torch.nn.Linear(8, 2)
    extra = ckpt.load_checkpoint(p, m2, torch.optim.SGD(m2.parameters(), lr=0.01))
    assert extra == {"epoch": 1}
    for a, b in zip(m.parameters(), m2.parameters()):
        assert torch.equal(a, b)


# ---- same-node P2P guard (mpi_operator_amd.parallel.p2p) ----

def _worker_p2p_groups(rank, world, port, q):
    try:
        _dist_env(rank, world, port)
        import torch.distributed as dist
        dist.init_process_group("gloo")
        from mpi_operator_amd.parallel import p2p
        groups = p2p.same_node_groups()
        # both CPU ranks run in one container -> one node group with both
        assert sorted(groups[0]) == [0, 1], groups
        # no CUDA here: verify_p2p is a no-op report
        rep = p2p.verify_p2p()
        assert rep["checked"] in (False, True)
        q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        q.put((rank, f"err {e}"))
    finally:
        import torch.distributed as dist
        if dist.is_initialized():
            dist.destroy_process_group()


def test_p2p_same_node_groups():
    _run_workers(_worker_p2p_groups, world=2, port=29741)


def test_p2p_static_preconditions(monkeypatch):
    from mpi_operator_amd.parallel import p2p
    monkeypatch.setenv("HSA_ENABLE_IPC_MODE_LEGACY", "1")
    probs = p2p.static_preconditions()
    assert any("HSA_ENABLE_IPC_MODE_LEGACY" in p for p in probs)
    monkeypatch.setenv("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    monkeypatch.setenv("NCCL_P2P_DISABLE", "1")
    probs = p2p.static_preconditions()
    assert any("NCCL_P2P_DISABLE" in p for p in probs)


# ---- Adasum reduction (Horovod's hvd.Adasum parity) ----

def test_adasum_combine_orthogonal_sums():
    """Orthogonal gradients add exactly; identical gradients average."""
    import torch
    from mpi_operator_amd.parallel.distributed_optimizer import _adasum_combine
    a = torch.tensor([1.0, 0.0])
    b = torch.tensor([0.0, 2.0])
    assert torch.allclose(_adasum_combine(a, b), torch.tensor([1.0, 2.0]))
    c = torch.tensor([3.0, 4.0])
    assert torch.allclose(_adasum_combine(c, c.clone()), c)  # avg of equals


def _worker_adasum(rank, world, port, q):
    try:
        _dist_env(rank, world, port)
        import torch
        import torch.distributed as dist
        dist.init_process_group("gloo")
        from mpi_operator_amd import parallel as hvd
        from mpi_operator_amd.parallel import DistributedOptimizer
        torch.manual_seed(100 + rank)
        m = torch.nn.Linear(8, 4)
        opt = DistributedOptimizer(torch.optim.SGD(m.parameters(), lr=0.1),
                                   op="adasum")
        x = torch.randn(4, 8)
        m(x).sum().backward()
        # fire + reduce
        opt.synchronize()
        # result must be identical on all ranks (adasum is symmetric)
        flat = torch.cat([p.grad.flatten() for p in m.parameters()])
        gathered = [torch.empty_like(flat) for _ in range(world)]
        dist.all_gather(gathered, flat)
        assert torch.allclose(gathered[0], gathered[1], atol=1e-6)
        q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        q.put((rank, f"err {e}"))
    finally:
        import torch.distributed as dist
        if dist.is_initialized():
            dist.destroy_process_group()


def test_adasum_two_rank_symmetry():
    _run_workers(_worker_adasum, world=2, port=29745)
