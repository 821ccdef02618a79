import torch

from simple_tip_amd.models import Cifar10CNN, ImdbTransformer, MnistCNN, ResNet20


def test_mnist_cnn_shapes():
    m = MnistCNN().eval()
    x = torch.randn(4, 1, 28, 28)
    taps, logits = m.forward_taps(x, [0, 1, 2, 3])
    assert logits.shape == (4, 10)
    assert [tuple(t.shape) for t in taps] == [
        (4, 32, 26, 26),
        (4, 32, 13, 13),
        (4, 64, 11, 11),
        (4, 64, 5, 5),
    ]
    # SA tap (layer 3) flattens to the reference's 1600 ATs
    assert taps[3].reshape(4, -1).shape[1] == 1600
    assert m.has_dropout()


def test_cifar_cnn_shapes():
    m = Cifar10CNN().eval()
    x = torch.randn(2, 3, 32, 32)
    taps, logits = m.forward_taps(x, [3])
    assert logits.shape == (2, 10)
    # SA tap = pool2: 6x6x64 = 2304 ATs (reference SURVEY)
    assert taps[0].reshape(2, -1).shape[1] == 2304
    assert not m.has_dropout()  # no VR for cifar10 (reference parity)


def test_resnet20_shapes():
    m = ResNet20().eval()
    x = torch.randn(2, 3, 32, 32)
    taps, logits = m.forward_taps(x, ResNet20.sa_layers)
    assert logits.shape == (2, 10)
    assert taps[0].reshape(2, -1).shape[1] == 8 * 8 * 64  # 4096-wide SA tap
    n_params = sum(p.numel() for p in m.parameters())
    assert 0.2e6 < n_params < 0.35e6  # ResNet-20 is ~0.27M params


def test_imdb_transformer_shapes():
    m = ImdbTransformer().eval()
    x = torch.randint(0, 2000, (3, 100))
    taps, logits = m.forward_taps(x, [3, 5])
    assert logits.shape == (3, 2)
    assert tuple(taps[0].shape) == (3, 32)  # GAP output
    assert tuple(taps[1].shape) == (3, 20)  # dense20 (SA tap)
    assert m.has_dropout()


def test_forward_equals_forward_taps():
    m = MnistCNN().eval()
    x = torch.randn(4, 1, 28, 28)
    with torch.no_grad():
        direct = m(x)
    _, tapped = m.forward_taps(x, [0])
    assert torch.allclose(direct, tapped)


def test_bn_folding_numerics():
    from simple_tip_amd.models.fuse import fold_bn_inference

    torch.manual_seed(3)
    m = ResNet20()
    for mod in m.modules():
        if isinstance(mod, torch.nn.BatchNorm2d):
            mod.running_mean.normal_(0, 0.3)
            mod.running_var.uniform_(0.5, 2.0)
            mod.weight.data.uniform_(0.5, 1.5)
            mod.bias.data.normal_(0, 0.3)
    m.eval()
    folded = fold_bn_inference(m)
    # no BatchNorm modules remain
    assert not any(
        isinstance(mod, torch.nn.BatchNorm2d) for mod in folded.modules()
    )
    x = torch.randn(4, 3, 32, 32)
    with torch.no_grad():
        a = m(x)
        b = folded(x)
    assert torch.allclose(a, b, rtol=1e-4, atol=1e-5)


def test_pad_stem_channels_identity():
    from simple_tip_amd.models.fuse import fold_bn_inference, pad_stem_channels

    torch.manual_seed(4)
    folded = fold_bn_inference(ResNet20()).eval()
    x = torch.randn(2, 3, 32, 32)
    with torch.no_grad():
        ref = folded(x)
    pad_stem_channels(folded, 4)
    x4 = torch.cat([x, torch.zeros(2, 1, 32, 32)], dim=1)
    with torch.no_grad():
        out = folded(x4)
    assert torch.allclose(ref, out, rtol=1e-5, atol=1e-6)
