"""IMDB text pipeline: synthetic texts -> TextCorruptor -> Tokenizer ->
padded sequences -> transformer forward (the reference's IMDB-C path)."""

import numpy as np
import torch

from simple_tip_amd.studies.imdb import ImdbCaseStudy, INPUT_MAXLEN, VOCAB_SIZE


def test_text_pipeline_end_to_end():
    study = ImdbCaseStudy(scale=0.01, device=torch.device("cpu"))
    train, nominal, ood = study.load_text_datasets(n_train=200, n_test=64)
    (tx, ty), (nx, ny), (ox, oy) = train, nominal, ood
    assert tx.shape == (200, INPUT_MAXLEN) and tx.dtype == np.int64
    assert nx.shape == (64, INPUT_MAXLEN)
    assert ox.shape == (128, INPUT_MAXLEN)  # nominal ++ corrupted
    assert tx.max() < VOCAB_SIZE and tx.min() >= 0
    # corrupted tokens differ from the nominal distribution but remain valid
    assert ox.max() < VOCAB_SIZE

    model = study.build_model().eval()
    with torch.no_grad():
        logits = model(torch.from_numpy(nx[:8]))
    assert logits.shape == (8, 2)


def test_text_pipeline_deterministic():
    study = ImdbCaseStudy(scale=0.01, device=torch.device("cpu"))
    a = study.load_text_datasets(n_train=100, n_test=32)
    b = study.load_text_datasets(n_train=100, n_test=32)
    assert np.array_equal(a[0][0], b[0][0])
    assert np.array_equal(a[2][0], b[2][0])
