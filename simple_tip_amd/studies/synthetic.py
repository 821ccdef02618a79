"""Deterministic synthetic datasets with class structure.

There is no network access for MNIST/CIFAR/IMDB downloads (BASELINE.json:
"on synthetic data / random-init weights"), so each case study generates a
class-conditional synthetic stand-in with the reference dataset's exact
shapes and split sizes. The generator is seeded per (study, split): models
genuinely learn the class structure (accuracy well above chance, but with
real misclassifications for APFD to rank), and the OOD split follows the
reference recipe — corrupted data concatenated with the nominal test set and
shuffled with seed 0 (reference case_study_mnist.py:31-47,161-165).
"""

import hashlib
from typing import Tuple

import numpy as np


def _rng(study: str, split: str, seed: int = 0) -> np.random.RandomState:
    h = int(hashlib.md5(f"{study}:{split}:{seed}".encode()).hexdigest()[:8], 16)
    return np.random.RandomState(h)


def image_class_templates(study: str, shape: Tuple[int, ...], num_classes: int) -> np.ndarray:
    """Smooth per-class template images (fixed per study)."""
    rng = _rng(study, "templates")
    t = rng.randn(num_classes, *shape).astype(np.float32)
    # cheap low-pass so templates have spatial structure convs can pick up
    for _ in range(2):
        t = (
            t
            + np.roll(t, 1, axis=-1)
            + np.roll(t, -1, axis=-1)
            + np.roll(t, 1, axis=-2)
            + np.roll(t, -1, axis=-2)
        ) / 5.0
    return t


def synthetic_images(
    study: str,
    split: str,
    n: int,
    shape: Tuple[int, ...],
    num_classes: int,
    noise: float = 0.8,
    ambiguous: float = 0.08,
) -> Tuple[np.ndarray, np.ndarray]:
    """(x, y): class template + Gaussian noise, scaled to roughly [0, 1].

    An ``ambiguous`` fraction of samples blends the labeled class's template
    ~50/50 with another class's, making them irreducibly hard: a trained
    model misclassifies about half of them (a few % overall), so nominal
    test sets contain real faults and nominal APFD is well-defined — the
    reference's real datasets have this property intrinsically
    (reference eval_apfd_table.py:111-131 expects a
    populated nominal column)."""
    rng = _rng(study, split)
    y = rng.randint(0, num_classes, size=n)
    templates = image_class_templates(study, shape, num_classes)
    base = templates[y].copy()
    if ambiguous > 0 and num_classes > 1:
        m = rng.rand(n) < ambiguous
        alt = (y + rng.randint(1, num_classes, size=n)) % num_classes
        w = (0.45 + 0.10 * rng.rand(n)).astype(np.float32)
        wm = w[m].reshape((-1,) + (1,) * len(shape))
        base[m] = (1.0 - wm) * base[m] + wm * templates[alt[m]]
    x = base + noise * rng.randn(n, *shape).astype(np.float32)
    x = (x - x.min()) / (x.max() - x.min() + 1e-8)
    return x.astype(np.float32), y.astype(np.int64)


def corrupt_images(study: str, x: np.ndarray, severity: float = 0.5) -> np.ndarray:
    """Corruption stand-in (MNIST-C / CIFAR-10-C role): extra noise +
    contrast shift + partial occlusion, deterministic per study."""
    rng = _rng(study, "corruption")
    out = x.copy()
    out += severity * rng.randn(*x.shape).astype(np.float32)
    out *= 1.0 - 0.3 * severity
    # occlude a random square per image
    h, w = x.shape[-2], x.shape[-1]
    size = max(1, int(min(h, w) * 0.25 * severity * 2))
    ys = rng.randint(0, h - size + 1, size=x.shape[0])
    xs = rng.randint(0, w - size + 1, size=x.shape[0])
    for i in range(x.shape[0]):
        out[i, ..., ys[i] : ys[i] + size, xs[i] : xs[i] + size] = 0.0
    return np.clip(out, 0.0, 1.0).astype(np.float32)


def make_ood_split(
    nominal_x: np.ndarray, nominal_y: np.ndarray, corrupted_x: np.ndarray, corrupted_y: np.ndarray
) -> Tuple[np.ndarray, np.ndarray]:
    """Reference OOD recipe: concat nominal + corrupted, shuffle seed 0."""
    x = np.concatenate([nominal_x, corrupted_x])
    y = np.concatenate([nominal_y, corrupted_y])
    idx = np.random.RandomState(0).permutation(len(x))
    return x[idx], y[idx]


def synthetic_tokens(
    study: str,
    split: str,
    n: int,
    seq_len: int,
    vocab_size: int,
    num_classes: int,
) -> Tuple[np.ndarray, np.ndarray]:
    """Class-conditional token sequences (IMDB stand-in).

    Each class has its own Zipf-ish token distribution so the transformer
    can learn sentiment-like structure.
    """
    rng = _rng(study, split)
    y = rng.randint(0, num_classes, size=n)
    base = 1.0 / (np.arange(vocab_size) + 10.0)
    class_boost = _rng(study, "token-classes").rand(num_classes, vocab_size) ** 4
    x = np.empty((n, seq_len), dtype=np.int64)
    class_probs = np.empty((num_classes, vocab_size))
    for c in range(num_classes):
        probs = base * (1.0 + 8.0 * class_boost[c])
        class_probs[c] = probs / probs.sum()
        sel = y == c
        x[sel] = rng.choice(vocab_size, size=(int(sel.sum()), seq_len), p=class_probs[c])
    # ambiguous fraction: tokens drawn from a ~50/50 two-class mixture, so
    # trained models have an irreducible nominal error (see synthetic_images)
    ambiguous = 0.08
    if num_classes > 1:
        m = np.nonzero(rng.rand(n) < ambiguous)[0]
        alt = (y[m] + rng.randint(1, num_classes, size=m.shape[0])) % num_classes
        w = 0.45 + 0.10 * rng.rand(m.shape[0])
        for j, i in enumerate(m):
            probs = (1.0 - w[j]) * class_probs[y[i]] + w[j] * class_probs[alt[j]]
            x[i] = rng.choice(vocab_size, size=seq_len, p=probs)
    return x, y.astype(np.int64)


def corrupt_tokens(study: str, x: np.ndarray, vocab_size: int, severity: float = 0.5) -> np.ndarray:
    """Token-level corruption (IMDB-C stand-in): random substitutions."""
    rng = _rng(study, "token-corruption")
    out = x.copy()
    mask = rng.rand(*x.shape) < severity
    out[mask] = rng.randint(0, vocab_size, size=int(mask.sum()))
    return out
