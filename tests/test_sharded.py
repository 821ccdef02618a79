"""Sharded partial-reduction tests: single-process semantics plus gloo
world_size=2 equivalence with the unsharded path."""

import multiprocessing as mp
import os

import numpy as np
import pytest
import torch

from simple_tip_amd import ops
from simple_tip_amd.parallel import sharded


def test_single_process_passthrough():
    a = torch.randn(10, 8)
    b = torch.randn(30, 8)
    d, i = sharded.sharded_rowmin_l2(a, b, 0)
    d0, i0 = ops.rowmin_l2(a, b)
    assert torch.equal(d, d0) and torch.equal(i, i0)
    l = sharded.sharded_kde_logsumexp(a, b)
    assert torch.allclose(l, ops.kde_logsumexp(a, b))


def _worker(rank, world, port, q):
    try:
        os.environ.update(
            MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
            WORLD_SIZE=str(world), RANK=str(rank), LOCAL_RANK=str(rank),
        )
        import torch.distributed as dist

        from simple_tip_amd.parallel import dist as pdist
        from simple_tip_amd.parallel import sharded as sh

        pdist.init_from_env(backend="gloo")
        torch.manual_seed(0)  # same full data on every rank
        test = torch.randn(23, 6, dtype=torch.float64)
        train = torch.randn(57, 6, dtype=torch.float64)

        local, off = sh.shard_rows(train)
        d, i = sh.sharded_rowmin_l2(test, local, off)
        d_ref, i_ref = ops.rowmin_l2(test, train)
        assert torch.allclose(d, d_ref, atol=1e-12), "sharded min mismatch"
        assert torch.equal(i, i_ref), "sharded argmin mismatch"

        lse = sh.sharded_kde_logsumexp(test, local)
        lse_ref = ops.kde_logsumexp(test, train)
        assert torch.allclose(lse, lse_ref, atol=1e-10)

        # welford merge across shards == global moments
        full = torch.randn(41, 5, dtype=torch.float64)
        loc, _ = sh.shard_rows(full)
        c, m, m2 = (
            float(loc.shape[0]),
            loc.mean(dim=0),
            ((loc - loc.mean(dim=0)) ** 2).sum(dim=0),
        )
        tc, tm, tm2 = sh.allreduce_welford(c, m, m2)
        assert tc == 41
        assert torch.allclose(tm, full.mean(dim=0), atol=1e-12)
        assert torch.allclose(
            tm2 / (tc - 1), full.var(dim=0, unbiased=True), atol=1e-12
        )

        mins, maxs = sh.allreduce_minmax(
            loc.min(dim=0).values.clone(), loc.max(dim=0).values.clone()
        )
        assert torch.equal(mins, full.min(dim=0).values)
        assert torch.equal(maxs, full.max(dim=0).values)

        dist.destroy_process_group()
        q.put((rank, "ok"))
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"fail: {e!r}"))


def test_gloo_world2_sharded_matches_unsharded():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = int(np.random.RandomState(os.getpid() + 1).randint(20000, 40000))
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=180) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"


def _ddp_worker(rank, world, port, q):
    try:
        os.environ.update(
            MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
            WORLD_SIZE=str(world), RANK=str(rank), LOCAL_RANK=str(rank),
        )
        import torch.distributed as dist

        from simple_tip_amd.parallel import dist as pdist
        from simple_tip_amd.models import MnistCNN
        from simple_tip_amd.studies.base import train_classifier

        pdist.init_from_env(backend="gloo")
        rng = np.random.RandomState(0)
        x = rng.rand(64, 1, 28, 28).astype(np.float32)
        y = rng.randint(0, 10, 64)
        torch.manual_seed(0)
        model = train_classifier(
            MnistCNN(), x, y, epochs=1, batch_size=16,
            device=torch.device("cpu"), seed=0,
        )
        # DDP keeps replicas in sync: same weights on both ranks
        w = model.layers[6].weight.detach().flatten()[:10]
        ws = [torch.empty_like(w) for _ in range(world)]
        dist.all_gather(ws, w)
        assert torch.allclose(ws[0], ws[1], atol=1e-6)
        dist.destroy_process_group()
        q.put((rank, "ok"))
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"fail: {e!r}"))


def test_gloo_world2_ddp_training():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = int(np.random.RandomState(os.getpid() + 2).randint(20000, 40000))
    procs = [ctx.Process(target=_ddp_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=240) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"


def test_fold_rowmin_partials_property():
    """hypothesis: for ANY split of the train rows into contiguous shards,
    the rank-ordered fold of per-shard (min, global argmin) equals the dense
    single-pass result — including the lowest-index tie rule."""
    from hypothesis import given, settings, strategies as st

    from simple_tip_amd.parallel.sharded import fold_rowmin_partials

    @settings(max_examples=60, deadline=None)
    @given(
        n=st.integers(4, 60),
        m=st.integers(1, 12),
        world=st.integers(1, 5),
        seed=st.integers(0, 10_000),
        quantize=st.booleans(),
    )
    def check(n, m, world, seed, quantize):
        g = torch.Generator().manual_seed(seed)
        test = torch.randn(m, 4, generator=g).double()
        train = torch.randn(n, 4, generator=g).double()
        if quantize:  # force exact distance ties to exercise the tie rule
            test = test.round()
            train = train.round()
        dense_d, dense_i = ops.rowmin_l2(test, train)
        # split into `world` contiguous shards (possibly empty)
        cuts = sorted(
            int(x) for x in torch.randint(0, n + 1, (world - 1,), generator=g)
        )
        bounds = [0] + cuts + [n]
        dists, idxs = [], []
        for r in range(world):
            lo, hi = bounds[r], bounds[r + 1]
            if hi == lo:
                dists.append(torch.full((m,), float("inf")).double())
                idxs.append(torch.zeros(m, dtype=torch.int64))
            else:
                d, i = ops.rowmin_l2(test, train[lo:hi])
                dists.append(d)
                idxs.append(i + lo)
        fd, fi = fold_rowmin_partials(dists, idxs)
        assert torch.equal(fd, dense_d)
        assert torch.equal(fi, dense_i)

    check()
