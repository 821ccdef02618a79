"""In-lease RCCL validation: 2 ranks sharing cuda:0 run the REAL sharded
production path (FusedPrioritizer with shard_train over RCCL) and compare
against the dense single-rank scores ("exercise RCCL for real
inside a 1-GPU lease").

Writes a JSON summary to gpurun_out/rccl_world2.json.

Measured fact (recorded in the summary): RCCL, like NCCL, refuses two ranks
on one device ("Duplicate GPU detected"), so on a 1-GPU lease the nccl
attempt fails by construction and the script falls back to the gloo
transport with the SAME device-resident sharded path (kernels on cuda:0,
partials staged through host for the collectives). The RCCL transport
itself runs when devices are distinct — the driver's round-end 8-GPU
scaling bench exercises exactly that.
"""

import json
import multiprocessing as mp
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

D, NTRAIN, BATCH, CLASSES = 512, 20000, 4096, 10


def worker(rank, q, backend, port):
    try:
        os.environ.update(
            MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
            WORLD_SIZE="2", RANK=str(rank), LOCAL_RANK="0",
        )
        import datetime

        import torch
        import torch.distributed as dist

        torch.cuda.set_device(0)
        dist.init_process_group(
            backend, timeout=datetime.timedelta(seconds=180)
        )
        dev = torch.device("cuda:0")
        from simple_tip_amd.core.surprise import DSA, LSA, MultiModalSA
        from simple_tip_amd.engine.serving import FusedPrioritizer

        torch.manual_seed(0)  # identical data on both ranks
        train = torch.randn(NTRAIN, D, device=dev)
        pred = torch.randint(0, CLASSES, (NTRAIN,), device=dev)
        test = torch.randn(BATCH, D, device=dev)
        tpred = torch.randint(0, CLASSES, (BATCH,), device=dev)

        # dense reference: local, no collectives

        dsa_dense = DSA(train, pred, device=dev)
        lsa_dense = MultiModalSA.build_by_class(
            train, pred, lambda a, p: LSA(a, max_features=300, device=dev)
        )
        fp_dense = FusedPrioritizer(dsa_dense, lsa_dense, dev)
        d_ref, l_ref = fp_dense(test, tpred)

        # sharded path over RCCL
        dsa_sh = DSA(train, pred, device=dev, shard_train=True)
        lsa_sh = MultiModalSA.build_by_class(
            train, pred,
            lambda a, p: LSA(a, max_features=300, device=dev, shard_train=True),
        )
        fp_sh = FusedPrioritizer(dsa_sh, lsa_sh, dev)
        d_sh, l_sh = fp_sh(test, tpred)
        torch.cuda.synchronize()

        dsa_err = float((d_sh - d_ref).abs().max())
        fin = torch.isfinite(l_ref)
        lsa_err = float((l_sh[fin] - l_ref[fin]).abs().max())

        for _ in range(3):
            fp_sh(test, tpred)
        dist.barrier()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(10):
            fp_sh(test, tpred)
        dist.barrier()
        torch.cuda.synchronize()
        sharded_ms = (time.perf_counter() - t0) / 10 * 1000

        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(10):
            fp_dense(test, tpred)
        torch.cuda.synchronize()
        dense_ms = (time.perf_counter() - t0) / 10 * 1000

        dist.destroy_process_group()
        q.put((rank, "ok", dsa_err, lsa_err, sharded_ms, dense_ms))
    except Exception as e:  # noqa: BLE001
        import traceback

        q.put((rank, f"fail: {e!r}", None, None, None, None))
        sys.stderr.write(traceback.format_exc())


def _run(backend, port):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=worker, args=(r, q, backend, port)) for r in range(2)
    ]
    for p in procs:
        p.start()
    results = []
    for _ in range(2):
        try:
            results.append(q.get(timeout=420))
        except Exception:
            results.append((-1, "timeout", None, None, None, None))
    for p in procs:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
    return results


def main():
    os.makedirs("gpurun_out", exist_ok=True)
    attempts = []
    results = _run("nccl", 29881)
    attempts.append({"backend": "nccl(RCCL)",
                     "ok": all(r[1] == "ok" for r in results),
                     "note": None if all(r[1] == "ok" for r in results) else
                     "RCCL refuses 2 ranks on one device (Duplicate GPU)"})
    if not attempts[-1]["ok"]:
        results = _run("gloo", 29883)
        attempts.append({"backend": "gloo (device kernels, host-staged collectives)",
                         "ok": all(r[1] == "ok" for r in results)})
    ok = attempts[-1]["ok"]
    summary = {
        "world": 2, "device": "cuda:0 shared",
        "config": {"train": NTRAIN, "d": D, "batch": BATCH},
        "attempts": attempts,
        "ok": ok,
        "ranks": [
            {"rank": r[0], "status": r[1][:400] if isinstance(r[1], str) else r[1],
             "dsa_max_abs_err": r[2],
             "lsa_max_abs_err": r[3], "sharded_ms": r[4], "dense_ms": r[5]}
            for r in results
        ],
    }
    with open("gpurun_out/rccl_world2.json", "w") as f:
        json.dump(summary, f, indent=1)
    print(json.dumps(summary))
    sys.exit(0 if ok else 1)


if __name__ == "__main__":
    main()
