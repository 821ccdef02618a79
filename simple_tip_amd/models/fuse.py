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
