"""Profiling driver for the non-pairwise HIP kernels (coverage profiles,
CAM iteration, softmax scores, bucketize, popcount) at engine-like shapes.

Run under rocprofv3 for the per-kernel stats / PMC evidence the round-1
review asked for:

  cd /tmp && export TMPDIR=/tmp
  rocprofv3 --stats --kernel-trace -d OUT -o small -- \
      python /root/repo/scripts/perf_small_ops.py
  rocprofv3 --pmc SQ_WAVE_CYCLES SQ_INSTS_VALU SQ_INSTS_LDS --kernel-trace \
      -d OUT2 -o smallpmc -- python /root/repo/scripts/perf_small_ops.py

Also prints wall-per-op timings (hipEvent-bracketed) on its own.
"""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from simple_tip_amd import ops
from simple_tip_amd.core.bitmap import BitProfile

DEV = torch.device("cuda:0")
N = 20000          # test inputs
KNEUR = 4160       # neurons (tap width, 64-multiple)
ITERS = 20


def timed(name, fn):
    fn()  # warm
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(ITERS):
        out = fn()
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / ITERS * 1000
    print(f"{name:24s} {ms:8.3f} ms/iter")
    return out


def main():
    assert torch.cuda.is_available() and ops.hip_available()
    torch.manual_seed(0)
    acts = torch.randn(N, KNEUR, device=DEV)
    layer_acts = [acts[:, :2048].contiguous(), acts[:, 2048:].contiguous()]
    mins = acts.min(dim=0).values - 0.1
    maxs = acts.max(dim=0).values - 0.5  # some out-of-range on purpose
    probs = torch.softmax(torch.randn(N, 10, device=DEV), dim=1)
    sa_vals = torch.rand(N, device=DEV).double()
    thresholds = torch.linspace(0, 1, 1001, dtype=torch.float64)

    timed("nac_profile", lambda: ops.nac_profile(acts, 0.5))
    timed("snac_profile", lambda: ops.snac_profile(acts, maxs))
    timed("nbc_profile", lambda: ops.nbc_profile(acts, mins, maxs))
    timed("kmnc_profile", lambda: ops.kmnc_profile(acts, mins, maxs, 2))
    timed("tknc_profile k=3", lambda: ops.tknc_profile(layer_acts, 3))
    timed("softmax_scores", lambda: ops.softmax_uncertainties(probs))
    words = timed(
        "bucketize_profile",
        lambda: ops.bucketize_profile(sa_vals, thresholds.to(DEV)),
    )
    prof = BitProfile(words, 1000)
    timed("popcount_rows", lambda: ops.popcount_rows(prof.words))

    scores = prof.popcount().float()
    ops.cam_order(scores, prof.words, prof.nbits)  # warm torch sort modules
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    order = ops.cam_order(scores, prof.words, prof.nbits)
    torch.cuda.synchronize()
    print(f"{'cam_order (warmed)':24s} {(time.perf_counter()-t0)*1000:8.3f} ms "
          f"({int(order.shape[0])} rows)")


if __name__ == "__main__":
    main()


def cam_diag():
    """Pick-count/time scaling of cam_greedy itself."""
    from simple_tip_amd.ops import _load_compiled

    ext = _load_compiled()
    for nbits, density in ((1000, 1), (1000, 8), (8000, 1)):
        vals = torch.rand(N, device=DEV).double()
        thr = torch.linspace(0, 1, nbits + 1, dtype=torch.float64).to(DEV)
        words = ops.bucketize_profile(vals, thr)
        if density > 1:  # OR together several shifted profiles
            w = words.clone()
            for s in range(1, density):
                w = w | words.roll(s * 37, dims=0)
            words = w
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        picked = ext.cam_greedy(words.contiguous(), nbits)
        torch.cuda.synchronize()
        ms = (time.perf_counter() - t0) * 1000
        print(f"cam_greedy nbits={nbits} density~{density}: {ms:8.2f} ms, "
              f"picks={picked.numel()}, ms/pick={ms/max(1,picked.numel()):.4f}")


if os.environ.get("TIP_CAM_DIAG") == "1":
    cam_diag()
