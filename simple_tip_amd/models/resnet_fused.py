"""Fused ResNet-20 inference runner (hand-written CDNA4 kernels).

Wraps ops/hip/resnet_fused.hip: one kernel launch per residual block (input
plane staged to LDS with halo, conv1 intermediate never leaves the CU),
bf16 NHWC, MFMA 16x16x32. Weights come from a BN-folded :class:`ResNet20`
(models/fuse.py) and are pre-packed on the host into the MFMA B-fragment
order [cout_tiles][ksteps][64 lanes][8].

Feature-order note: activations are NHWC, so the 4096-wide stage-3 AT is
HWC-flattened (the torch path flattens CHW). L2/KDE/Mahalanobis scores are
invariant under any FIXED feature permutation; train and test ATs must
simply come from the same path.
"""

from typing import List, Tuple

import torch

from ..ops import _load_compiled
from .cnn import ResNet20, _BasicBlock


def _pack_conv(w: torch.Tensor, taps: int) -> torch.Tensor:
    """Pack an OIHW conv weight into MFMA B-fragments.

    Returns int-represented bf16 tensor [cout_tiles, ksteps, 64, 8] with
    wpack[ct, ks, l, e] = W2[ks*32 + (l>>4)*8 + e][ct*16 + (l&15)] where
    W2[k][cout], k = tap*Cin + ci (taps row-major dy,dx)."""
    cout, cin = w.shape[0], w.shape[1]
    k_total = taps * cin
    ksteps = (k_total + 31) // 32
    # W2: [K, cout]
    if taps == 9:
        w2 = w.permute(2, 3, 1, 0).reshape(k_total, cout)  # (dy,dx,ci) x cout
    else:
        w2 = w.reshape(cout, cin).t()  # 1x1: [cin, cout]
    w2 = w2.float()
    padded = torch.zeros(ksteps * 32, cout, dtype=torch.float32)
    padded[:k_total] = w2
    lanes = torch.arange(64)
    g = lanes >> 4
    j = lanes & 15
    out = torch.zeros(cout // 16, ksteps, 64, 8, dtype=torch.float32)
    for ct in range(cout // 16):
        for ks in range(ksteps):
            for e in range(8):
                out[ct, ks, :, e] = padded[ks * 32 + g * 8 + e, ct * 16 + j]
    return out.to(torch.bfloat16).contiguous()


class FusedResNet20:
    """Inference-only fused forward; returns (stage3 ATs [B,4096], logits)."""

    def __init__(self, folded: ResNet20, device):
        ext = _load_compiled()
        self.ext = ext
        self.device = device
        layers = folded.layers
        # stem: Sequential(conv(+folded bias), relu); pad Cin 4->8 is NOT
        # needed here — we pad 3->8 directly (kernel uses CIN=8 units)
        stem_conv = layers[0][0]
        w = stem_conv.weight.detach().float()
        w8 = torch.zeros(16, 8, 3, 3)
        w8[:, : w.shape[1]] = w
        self.stem_w = _pack_conv(w8, 9).to(device)
        self.stem_b = stem_conv.bias.detach().float().to(device)

        self.blocks: List[Tuple] = []
        for layer in layers[1:10]:
            assert isinstance(layer, _BasicBlock)
            c_in = layer.conv1.in_channels
            c_out = layer.conv1.out_channels
            w1 = _pack_conv(layer.conv1.weight.detach().float(), 9).to(device)
            b1 = layer.conv1.bias.detach().float().to(device)
            w2 = _pack_conv(layer.conv2.weight.detach().float(), 9).to(device)
            b2 = layer.conv2.bias.detach().float().to(device)
            if isinstance(layer.shortcut, torch.nn.Sequential):
                sc = layer.shortcut[0]
                wsc = _pack_conv(sc.weight.detach().float(), 1).to(device)
                bsc = sc.bias.detach().float().to(device)
                variant = {16: 0, 32: 1}[c_in]
                self.blocks.append(("down", variant, w1, b1, w2, b2, wsc, bsc))
            else:
                variant = {16: 0, 32: 1, 64: 2}[c_in]
                self.blocks.append(("res", variant, w1, b1, w2, b2))

        pool_fc = layers[11]
        self.fc_w = pool_fc.weight.detach().float().to(device)
        self.fc_b = pool_fc.bias.detach().float().to(device)

    @torch.no_grad()
    def __call__(self, x_nchw: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        x = x_nchw.to(self.device, torch.float32)
        return self.forward_nhwc(x.permute(0, 2, 3, 1))

    @torch.no_grad()
    def forward_nhwc(self, x_nhwc: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        """Forward for NHWC inputs [B, 32, 32, C<=8] (skips the 125-MB
        strided NCHW->NHWC transpose on the hot path)."""
        b = x_nhwc.shape[0]
        x = x_nhwc.to(self.device)
        nhwc = torch.zeros(b, 32, 32, 8, device=self.device, dtype=torch.bfloat16)
        nhwc[..., : x.shape[-1]] = x.to(torch.bfloat16)
        cur = self.ext.resnet_stem(
            nhwc.reshape(b, -1).contiguous(), self.stem_w.reshape(-1, 8),
            self.stem_b,
        )
        for blk in self.blocks:
            if blk[0] == "res":
                _, variant, w1, b1, w2, b2 = blk
                cur = self.ext.resnet_block(
                    variant, cur, w1.reshape(-1, 8), b1, w2.reshape(-1, 8), b2
                )
            else:
                _, variant, w1, b1, w2, b2, wsc, bsc = blk
                cur = self.ext.resnet_down(
                    variant, cur, w1.reshape(-1, 8), b1, w2.reshape(-1, 8),
                    b2, wsc.reshape(-1, 8), bsc,
                )
        # cur: [B, 8*8*64] NHWC bf16 — the stage-3 AT tap. Returned in bf16:
        # the bf16 pairwise kernels consume it directly (zero-copy hot
        # path); fp32 consumers cast at their boundary.
        pooled = cur.reshape(b, 64, 64).float().mean(dim=1)  # avg over 8x8
        logits = pooled @ self.fc_w.t() + self.fc_b
        return cur, logits
