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
