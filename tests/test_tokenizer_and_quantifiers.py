import numpy as np
import pytest
import torch

from simple_tip_amd.core.quantifiers import QuantifierRegistry
from simple_tip_amd.utils.tokenizer import Tokenizer, pad_sequences


def test_tokenizer_rank_by_frequency():
    tok = Tokenizer(num_words=100).fit_on_texts(
        ["a a a b b c", "b c d!"]
    )
    assert tok.word_index["a"] == 1
    assert tok.word_index["b"] == 2
    seqs = tok.texts_to_sequences(["a b, unknown c"])
    assert seqs == [[1, 2, 3]]


def test_tokenizer_num_words_budget():
    tok = Tokenizer(num_words=3).fit_on_texts(["a a b b c"])
    # only indexes < num_words survive (keras semantics)
    assert tok.texts_to_sequences(["a b c"]) == [[1, 2]]


def test_pad_sequences_prepad_pretruncate():
    out = pad_sequences([[1, 2], [3, 4, 5, 6]], maxlen=3)
    assert out.tolist() == [[0, 1, 2], [4, 5, 6]]
    assert out.dtype == np.int64


def test_registry_aliases():
    q = QuantifierRegistry.find("custom::deep_gini")
    probs = torch.tensor([[0.5, 0.5]])
    pred, val = q.calculate(probs)
    assert val.item() == pytest.approx(0.5)
    assert QuantifierRegistry.find("softmax").is_confidence
    assert QuantifierRegistry.find("VR").takes_samples


def test_as_uncertainty_negates_confidence():
    q = QuantifierRegistry.find("softmax")
    probs = torch.tensor([[0.9, 0.1]])
    _, u = q.as_uncertainty(probs)
    assert u.item() == pytest.approx(-0.9)


def test_registry_matches_fused_kernel_names():
    from simple_tip_amd import ops

    probs = torch.softmax(torch.randn(8, 5), dim=1)
    fused = ops.softmax_uncertainties(probs)
    for name in fused:
        q = QuantifierRegistry.find(name)
        _, u = q.as_uncertainty(probs)
        assert torch.allclose(u, fused[name], atol=1e-6), name
