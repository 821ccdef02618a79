import numpy as np
import pytest

from simple_tip_amd.core.text_corruptor import (
    CorruptionType,
    CorruptionWeights,
    TextCorruptor,
    levenshtein_matrix,
)

CORPUS = [
    "the quick brown foxes jumped over seventeen lazy hounds yesterday",
    "quick thinking foxes always outrun lazier hounds every morning",
    "seventeen quick hounds chased brown foxes through autumn leaves",
    "every morning brings fresh autumn leaves and lazy thoughts",
] * 5


@pytest.fixture(scope="module")
def corruptor():
    return TextCorruptor(CORPUS, dict_size=50)


def test_levenshtein_matrix():
    words = ["kitten", "sitting", "kitchen", "mitten"]
    d = levenshtein_matrix(words)
    assert d[0, 1] == 3  # kitten -> sitting
    assert d[0, 3] == 1  # kitten -> mitten
    assert d[0, 0] == 0
    assert np.array_equal(d, d.T)


def test_dictionary_contract(corruptor):
    # length > 4, lowercase, no numbers, alphabetically sorted
    for w in corruptor.common_words:
        assert len(w) > 4 and w == w.lower() and not w.isdigit()
    assert corruptor.common_words == sorted(corruptor.common_words)
    assert "quick" in corruptor.common_words


def test_severity_zero_is_identity(corruptor):
    texts = ["quick brown foxes jumped"]
    assert corruptor.corrupt(texts, severity=0.0, seed=1) == texts


def test_determinism_independent_of_batch(corruptor):
    t1 = "seventeen lazy hounds chased quick foxes"
    t2 = "fresh autumn leaves every morning"
    alone = corruptor.corrupt([t1], severity=0.6, seed=3)[0]
    batched = corruptor.corrupt([t2, t1, t2], severity=0.6, seed=3)[1]
    assert alone == batched


def test_severity_monotonicity(corruptor):
    """Higher severity corrupts a superset of the words corrupted at a
    lower severity (the reference's documented contract)."""
    text = "seventeen lazy hounds chased quick brown foxes through autumn leaves"
    words = text.split()
    low = corruptor.corrupt([text], severity=0.3, seed=7)[0].split()
    high = corruptor.corrupt([text], severity=0.9, seed=7)[0].split()
    assert len(low) == len(high) == len(words)
    for orig, lo, hi in zip(words, low, high):
        if lo != orig:  # corrupted at low severity ...
            assert hi == lo  # ... must be identically corrupted at high


def test_corruption_changes_words(corruptor):
    text = "seventeen lazy hounds chased quick brown foxes through autumn leaves"
    out = corruptor.corrupt([text], severity=1.0, seed=11)[0]
    diff = sum(a != b for a, b in zip(text.split(), out.split()))
    assert diff >= 3


def test_typo_single_char():
    w = TextCorruptor._corrupt_typo("hello", seed=42)
    assert len(w) == 5
    assert sum(a != b for a, b in zip(w, "hello")) == 1


def test_synonym_dict_used():
    c = TextCorruptor(CORPUS, dict_size=50, synonyms={"quick": ["speedy"]})
    out = c._corrupt_word("quick", seed=5, ctype=CorruptionType.SYNONYM)
    assert out == "speedy"


def test_autocorrect_picks_near_word(corruptor):
    out = corruptor._corrupt_autocorrect("hounds", seed=1)
    if out != "hounds":
        assert out in corruptor.common_words


def test_weights_all_types_reachable(corruptor):
    text = " ".join(corruptor.common_words[:20])
    out = corruptor.corrupt(
        [text], severity=1.0, seed=0,
        weights=CorruptionWeights(typo_weight=1.0, autocomplete_weight=0,
                                  autocorrect_weight=0, synonym_weight=0),
    )[0]
    # pure-typo corruption: every corrupted word differs by one char
    for orig, new in zip(text.split(), out.split()):
        if new != orig:
            assert len(new) == len(orig)


def test_synonym_table_ships_in_tree():
    """SYNONYM (weight 0.35) must exercise its own path via the bundled
    thesaurus instead of silently degrading to AUTOCORRECT (VERDICT r01
    item 8; reference downloads WordNet, text_corruptor.py:412-446)."""
    from simple_tip_amd.core.synonyms_data import SYNONYMS
    from simple_tip_amd.core.text_corruptor import TextCorruptor

    assert len(SYNONYMS) >= 250
    assert all(s and all(isinstance(w, str) for w in s) for s in SYNONYMS.values())

    base = ["a wonderful story about a terrible movie"] * 5
    tc = TextCorruptor(base_dataset=base, dict_size=100)
    out = tc._corrupt_synonym("wonderful", seed=1)
    assert out in SYNONYMS["wonderful"]
    # determinism
    assert out == tc._corrupt_synonym("wonderful", seed=1)
    # unknown words still fall back to autocorrect semantics
    assert tc._corrupt_synonym("qqqqq", seed=1) == tc._corrupt_autocorrect("qqqqq", seed=1)
