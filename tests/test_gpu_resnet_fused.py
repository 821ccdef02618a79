"""Fused ResNet-20 inference kernels vs the torch reference (GPU)."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_mfma_probe_layout():
    """Verify the assumed v_mfma_f32_16x16x32_bf16 fragment layouts against
    a plain fp32 matmul of the bf16-rounded operands (asymmetric data)."""
    from simple_tip_amd.ops import _load_compiled

    ext = _load_compiled()
    rng = np.random.RandomState(0)
    a = torch.from_numpy(rng.randn(16, 32).astype(np.float32)).to(torch.bfloat16)
    b = torch.from_numpy(rng.randn(32, 16).astype(np.float32)).to(torch.bfloat16)
    d = ext.mfma_probe(a.cuda(), b.cuda()).cpu()
    want = a.float() @ b.float()
    assert torch.allclose(d, want, atol=1e-3, rtol=1e-3), (
        (d - want).abs().max()
    )


@pytest.fixture(scope="module")
def folded_pair():
    from simple_tip_amd.models import ResNet20
    from simple_tip_amd.models.fuse import fold_bn_inference

    torch.manual_seed(0)
    m = ResNet20()
    # randomize BN stats so folding is non-trivial
    for mod in m.modules():
        if isinstance(mod, torch.nn.BatchNorm2d):
            mod.running_mean.normal_(0, 0.2)
            mod.running_var.uniform_(0.5, 2.0)
            mod.weight.data.uniform_(0.5, 1.5)
            mod.bias.data.normal_(0, 0.2)
    folded = fold_bn_inference(m)
    return m, folded


def test_fused_forward_matches_torch(folded_pair):
    from simple_tip_amd.models.resnet_fused import FusedResNet20

    _, folded = folded_pair
    dev = torch.device("cuda:0")
    fused = FusedResNet20(folded, dev)
    torch.manual_seed(1)
    x = torch.randn(64, 3, 32, 32)

    ats, logits = fused(x)
    assert ats.shape == (64, 4096) and logits.shape == (64, 10)

    ref_model = folded.to(dev).eval()
    with torch.no_grad():
        taps, ref_logits = ref_model.forward_taps(x.to(dev), [9])
    ref_ats_nhwc = taps[0].permute(0, 2, 3, 1).reshape(64, -1)

    # both paths accumulate conv sums in fp32; bf16 rounding between layers
    # differs only in accumulation order
    a = ats.float()
    r = ref_ats_nhwc.float()
    denom = r.abs().mean().clamp_min(1e-3)
    rel = (a - r).abs().mean() / denom
    assert rel < 0.05, float(rel)

    agree = (logits.argmax(1) == ref_logits.argmax(1)).float().mean()
    assert agree > 0.95, float(agree)


def test_fused_forward_batch_invariance(folded_pair):
    from simple_tip_amd.models.resnet_fused import FusedResNet20

    _, folded = folded_pair
    dev = torch.device("cuda:0")
    fused = FusedResNet20(folded, dev)
    x = torch.randn(32, 3, 32, 32)
    a1, l1 = fused(x)
    a2, l2 = fused(x[:7])
    assert torch.equal(a1[:7], a2)
    assert torch.equal(l1[:7], l2)
