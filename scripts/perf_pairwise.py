"""Microbenchmark of the MFMA pairwise-distance kernel (gfx950).

Reports effective TFLOP/s (2*M*N*K flops) for the three epilogues at
workload-representative shapes. Run on the GPU box:
    python scripts/perf_pairwise.py
"""

import sys
import time

import torch

sys.path.insert(0, ".")
from simple_tip_amd.ops import hip_ops  # noqa: E402


def bench(fn, flops, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    return flops / dt / 1e12, dt * 1e3


def main():
    torch.manual_seed(0)
    shapes = [
        (4096, 4096, 4096),   # square reference
        (10240, 15000, 4096), # bench DSA hop-2 aggregate shape
        (1024, 1500, 4096),   # per-class DSA hop-1 shape
        (10240, 5000, 300),   # per-class LSA/KDE shape
    ]
    for m, n, k in shapes:
        a = torch.randn(m, k, device="cuda")
        b = torch.randn(n, k, device="cuda")
        fl = 2.0 * m * n * k
        tf_min, ms_min = bench(lambda: hip_ops.rowmin_l2(a, b), fl)
        tf_kde, ms_kde = bench(lambda: hip_ops.kde_logsumexp(a, b), fl)
        print(
            f"M={m} N={n} K={k}: rowmin {tf_min:7.1f} TF ({ms_min:7.2f} ms)  "
            f"kde {tf_kde:7.1f} TF ({ms_kde:7.2f} ms)",
            flush=True,
        )
        del a, b
    # torch matmul reference on the same gram shape (rocBLAS fp32)
    m, n, k = 4096, 4096, 4096
    a = torch.randn(m, k, device="cuda")
    b = torch.randn(n, k, device="cuda")
    tf, ms = bench(lambda: a @ b.t(), 2.0 * m * n * k)
    print(f"rocBLAS fp32 gram @4096^3: {tf:7.1f} TF ({ms:7.2f} ms)")


if __name__ == "__main__":
    main()
