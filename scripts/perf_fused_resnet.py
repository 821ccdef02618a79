"""Microbenchmark of the fused ResNet-20 forward (gfx950)."""

import sys
import time

import torch

sys.path.insert(0, ".")
from simple_tip_amd.models import ResNet20  # noqa: E402
from simple_tip_amd.models.fuse import fold_bn_inference  # noqa: E402
from simple_tip_amd.models.resnet_fused import FusedResNet20  # noqa: E402


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    b = 10240
    model = fold_bn_inference(ResNet20()).to(dev)
    fused = FusedResNet20(model, dev)
    x = torch.randn(b, 32, 32, 3, device=dev)
    xcpu = torch.randn(b, 32, 32, 3).pin_memory()

    print(f"full forward_nhwc: {timeit(lambda: fused.forward_nhwc(x)):.2f} ms")

    # stage-by-stage
    nhwc = torch.zeros(b, 32, 32, 8, device=dev, dtype=torch.bfloat16)

    def conv_in():
        nhwc[..., :3] = x.to(torch.bfloat16)

    print(f"  pad+cast: {timeit(conv_in):.2f} ms")
    xin = nhwc.reshape(b, -1).contiguous()
    stem_out = fused.ext.resnet_stem(xin, fused.stem_w.reshape(-1, 8), fused.stem_b)
    print(f"  stem: {timeit(lambda: fused.ext.resnet_stem(xin, fused.stem_w.reshape(-1, 8), fused.stem_b)):.2f} ms")
    cur = stem_out
    outs = []
    for bi, blk in enumerate(fused.blocks):
        cur_in = cur
        if blk[0] == "res":
            _, v, w1, b1, w2, b2 = blk
            fn = lambda: fused.ext.resnet_block(v, cur_in, w1.reshape(-1, 8), b1, w2.reshape(-1, 8), b2)
        else:
            _, v, w1, b1, w2, b2, wsc, bsc = blk
            fn = lambda: fused.ext.resnet_down(v, cur_in, w1.reshape(-1, 8), b1, w2.reshape(-1, 8), b2, wsc.reshape(-1, 8), bsc)
        print(f"  block{bi} ({blk[0]}): {timeit(fn):.2f} ms")
        cur = fn()
    print(f"  ats.float(): {timeit(lambda: cur.float()):.2f} ms")
    pooled = lambda: (cur.reshape(b, 64, 64).float().mean(dim=1) @ fused.fc_w.t() + fused.fc_b)
    print(f"  pool+fc: {timeit(pooled):.2f} ms")
    print(f"  H2D pinned 125MB: {timeit(lambda: x.copy_(xcpu, non_blocking=True)):.2f} ms")


if __name__ == "__main__":
    main()
