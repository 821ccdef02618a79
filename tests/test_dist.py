"""Multi-process (gloo, world_size 2) tests of the sharding helpers —
the CPU stand-in for the RCCL/xGMI path."""

import multiprocessing as mp
import os

import numpy as np
import pytest
import torch


def _worker(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["WORLD_SIZE"] = str(world)
        os.environ["RANK"] = str(rank)
        os.environ["LOCAL_RANK"] = str(rank)
        import torch.distributed as dist

        from simple_tip_amd.parallel import dist as pdist

        r, w, dev = pdist.init_from_env(backend="gloo")
        assert (r, w) == (rank, world)

        n_total = 11  # uneven: rank0 gets 6, rank1 gets 5
        s = pdist.shard_slice(n_total)
        full = torch.arange(n_total, dtype=torch.float32).unsqueeze(1) * 2.0
        local = full[s]
        gathered = pdist.allgather_rows(local, n_total)
        assert torch.equal(gathered, full)

        m = pdist.allreduce_max_scalar(float(rank) + 0.5, dev)
        assert m == world - 1 + 0.5

        dist.destroy_process_group()
        q.put((rank, "ok"))
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"fail: {e!r}"))


def test_gloo_world2_shard_and_gather():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = int(np.random.RandomState(os.getpid()).randint(20000, 40000))
    procs = [
        ctx.Process(target=_worker, args=(r, 2, port, q)) for r in range(2)
    ]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"


def test_shard_slice_single_process():
    from simple_tip_amd.parallel import shard_slice

    assert shard_slice(10, 0, 4) == slice(0, 3)
    assert shard_slice(10, 1, 4) == slice(3, 6)
    assert shard_slice(10, 3, 4) == slice(8, 10)
    # shards tile [0, n)
    covered = []
    for r in range(4):
        s = shard_slice(10, r, 4)
        covered.extend(range(s.start, s.stop))
    assert covered == list(range(10))
