"""Inference-path fusion: fold BatchNorm into convolutions.

Profiling (profiles/r01_bench_kernels.md) showed MIOpen BatchNorm
*inference* kernels dominating the AT-extraction forward (361 us per launch
on ResNet-20-sized tensors). At eval time BN is an affine map, so it folds
into the preceding conv's weights exactly — every BN launch disappears.
"""

import copy

import torch.nn as nn
from torch.nn.utils.fusion import fuse_conv_bn_eval

from .cnn import ResNet20, _BasicBlock


def pad_stem_channels(model, channels: int = 4):
    """Zero-pad the first conv's input channels (3 -> 4) in place.

    MIOpen has no NHWC bf16 igemm for 3-channel input and falls back to a
    ~1.25 ms naive conv (profiles/r01_bench_kernels.md); a zero input
    channel is mathematically identical and hits the fast path. Callers
    must pad the input tensor to match.
    """
    import torch

    first = None
    for mod in model.modules():
        if isinstance(mod, nn.Conv2d):
            first = mod
            break
    assert first is not None and first.in_channels <= channels
    if first.in_channels == channels:
        return model
    w = first.weight.data
    neww = torch.zeros(
        w.shape[0], channels, w.shape[2], w.shape[3],
        dtype=w.dtype, device=w.device,
    )
    neww[:, : w.shape[1]] = w
    first.weight = nn.Parameter(neww)
    first.in_channels = channels
    return model


def fold_bn_inference(model: ResNet20) -> ResNet20:
    """Return an eval-mode deep copy with all conv+BN pairs fused.

    Numerically identical (fp32 algebra on the folded weights) to the
    original eval forward; layer indexing is preserved so tap ids stay
    valid."""
    m = copy.deepcopy(model).eval()
    for i, layer in enumerate(m.layers):
        if isinstance(layer, _BasicBlock):
            layer.conv1 = fuse_conv_bn_eval(layer.conv1, layer.bn1)
            layer.bn1 = nn.Identity()
            layer.conv2 = fuse_conv_bn_eval(layer.conv2, layer.bn2)
            layer.bn2 = nn.Identity()
            if isinstance(layer.shortcut, nn.Sequential):
                layer.shortcut = nn.Sequential(
                    fuse_conv_bn_eval(layer.shortcut[0], layer.shortcut[1])
                )
        elif (
            isinstance(layer, nn.Sequential)
            and len(layer) == 3
            and isinstance(layer[0], nn.Conv2d)
            and isinstance(layer[1], nn.BatchNorm2d)
        ):
            m.layers[i] = nn.Sequential(
                fuse_conv_bn_eval(layer[0], layer[1]), layer[2]
            )
    return m
