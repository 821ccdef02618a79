"""Text corruption generator (IMDB-C role).

Capability parity with reference src/core/text_corruptor.py:92-508, with the
same behavioural contract:
- four corruption types — TYPO, SYNONYM, AUTOCOMPLETE, AUTOCORRECT — drawn
  with configurable weights (defaults .05/.35/.30/.30);
- fully deterministic per sentence: the seed is derived from an MD5 hash of
  the sentence text plus the user seed, so corruption is independent of the
  order/size of the dataset;
- severity monotonicity: a higher severity corrupts a superset of the words
  corrupted at lower severity (corruption types are drawn per word first,
  then a severity-sized prefix of a seeded shuffle selects which apply);
- the dictionary is the `dict_size` most common words (length > 4, not
  numeric) of a base corpus; AUTOCOMPLETE uses shared-prefix "start bags";
  AUTOCORRECT uses the all-pairs Levenshtein matrix (C++ threaded kernel in
  the extension; pure-python fallback for small dictionaries).

Deviation (no network egress in this environment): the reference downloads
a WordNet thesaurus for SYNONYM (text_corruptor.py:412-446). Here a compact
curated synonym table ships in-tree (synonyms_data.py) and is used by
default; a custom dict stays injectable. Words without synonyms fall back
to AUTOCORRECT (nearest dictionary word), which the reference also does.
"""

import collections
import dataclasses
import enum
import hashlib
import logging
import os
import pickle
import string
from typing import Dict, List, Optional

import numpy as np

logger = logging.getLogger(__name__)

MIN_COMMON_START = 3
MAX_COMMON_START = 5


class CorruptionType(enum.Enum):
    TYPO = 0
    SYNONYM = 1
    AUTOCOMPLETE = 2
    AUTOCORRECT = 3


@dataclasses.dataclass
class CorruptionWeights:
    typo_weight: float = 0.05
    autocomplete_weight: float = 0.30
    autocorrect_weight: float = 0.30
    synonym_weight: float = 0.35


def _hash_to_int(parts: List[str]) -> int:
    digest = hashlib.md5(" ".join(parts).encode("utf-8")).hexdigest()
    return int(digest, 16) % 1_000_000


def _levenshtein_py(a: str, b: str) -> int:
    dp = list(range(len(b) + 1))
    for r, ca in enumerate(a, 1):
        prev = dp[0]
        dp[0] = r
        for c, cb in enumerate(b, 1):
            cur = dp[c]
            dp[c] = min(prev + (ca != cb), dp[c] + 1, dp[c - 1] + 1)
            prev = cur
    return dp[-1]


def levenshtein_matrix(words: List[str]) -> np.ndarray:
    """All-pairs Levenshtein distances (uint8)."""
    try:
        from ..ops import _load_compiled

        ext = _load_compiled()
        return ext.levenshtein_matrix(list(words)).numpy()
    except Exception:
        n = len(words)
        out = np.zeros((n, n), dtype=np.uint8)
        for i in range(n):
            for j in range(i + 1, n):
                d = min(_levenshtein_py(words[i], words[j]), 255)
                out[i, j] = out[j, i] = d
        return out


class TextCorruptor:
    """Deterministic natural-looking corruption of text datasets."""

    def __init__(
        self,
        base_dataset: List[str],
        dict_size: int = 4000,
        cache_dir: Optional[str] = None,
        synonyms: Optional[Dict[str, List[str]]] = None,
    ):
        self.cache_dir = cache_dir
        if cache_dir:
            os.makedirs(cache_dir, exist_ok=True)
        self.common_words = self._extract_common_words(base_dataset, dict_size)
        self.word_index = {w: i for i, w in enumerate(self.common_words)}
        self.start_bags = self._word_start_bags()
        self.distances = self._calculate_distances()
        if synonyms is None:
            from .synonyms_data import SYNONYMS

            synonyms = SYNONYMS
        self.synonyms = synonyms

    # -- dictionary construction ----------------------------------------

    def _extract_common_words(self, base_dataset: List[str], size: int) -> List[str]:
        """The `size` most common lowercase words of length > 4 that are not
        numbers and contain a letter, sorted alphabetically."""
        if self.cache_dir:
            f = os.path.join(self.cache_dir, "common-words.pkl")
            if os.path.exists(f):
                with open(f, "rb") as fh:
                    return pickle.load(fh)
        words = [
            w.lower()
            for text in base_dataset
            for w in text.split()
        ]
        words = [
            w
            for w in words
            if len(w) > 4 and not w.isdigit() and any(c.isalpha() for c in w)
        ]
        chosen = sorted(dict(collections.Counter(words).most_common(size)).keys())
        if self.cache_dir:
            with open(os.path.join(self.cache_dir, "common-words.pkl"), "wb") as fh:
                pickle.dump(chosen, fh)
        return chosen

    def _word_start_bags(self) -> Dict[int, Dict[str, List[str]]]:
        """Per prefix length, bags of dictionary words sharing that prefix."""
        result = {}
        for k in range(MIN_COMMON_START, MAX_COMMON_START + 1):
            bag: Dict[str, List[str]] = {}
            for w in self.common_words:
                if len(w) >= k:
                    bag.setdefault(w[:k], []).append(w)
            result[k] = bag
        return result

    def _calculate_distances(self) -> np.ndarray:
        if self.cache_dir:
            f = os.path.join(self.cache_dir, "distances.npy")
            if os.path.exists(f):
                return np.load(f)
        d = levenshtein_matrix(self.common_words)
        if self.cache_dir:
            np.save(os.path.join(self.cache_dir, "distances.npy"), d)
        return d

    # -- corruption ------------------------------------------------------

    def corrupt(
        self,
        texts: List[str],
        severity: float,
        seed: int,
        weights: Optional[CorruptionWeights] = None,
    ) -> List[str]:
        """Corrupt a list of texts; see the module docstring for the
        determinism and severity-monotonicity guarantees."""
        assert 0.0 <= severity <= 1.0, "Severity must be between 0 and 1"
        weights = weights or CorruptionWeights()
        wvec = np.array(
            [
                weights.typo_weight,
                weights.autocomplete_weight,
                weights.autocorrect_weight,
                weights.synonym_weight,
            ]
        )
        wvec = wvec / wvec.sum()
        # index -> CorruptionType in the reference's draw order
        type_order = [
            CorruptionType.TYPO,
            CorruptionType.AUTOCOMPLETE,
            CorruptionType.AUTOCORRECT,
            CorruptionType.SYNONYM,
        ]

        out = []
        for text in texts:
            words = text.split()
            sentence_seed = _hash_to_int(words) + seed
            rng = np.random.default_rng(sentence_seed)
            ctypes = [type_order[rng.choice(4, p=wvec)] for _ in words]
            idx = np.arange(len(words))
            np.random.default_rng(sentence_seed).shuffle(idx)
            apply_set = set(idx[: round(len(words) * severity)].tolist())
            new_words = []
            for i, w in enumerate(words):
                if i not in apply_set or len(w) < 2:
                    new_words.append(w)
                else:
                    new_words.append(
                        self._corrupt_word(w, sentence_seed + i, ctypes[i])
                    )
            out.append(" ".join(new_words))
        return out

    def _corrupt_word(self, word: str, seed: int, ctype: CorruptionType) -> str:
        if ctype == CorruptionType.TYPO:
            return self._corrupt_typo(word, seed)
        if ctype == CorruptionType.SYNONYM:
            return self._corrupt_synonym(word, seed)
        if ctype == CorruptionType.AUTOCOMPLETE:
            return self._corrupt_autocomplete(word, seed)
        return self._corrupt_autocorrect(word, seed)

    @staticmethod
    def _corrupt_typo(word: str, seed: int) -> str:
        pos = seed % len(word)
        candidates = string.ascii_lowercase.replace(word[pos].lower(), "")
        typo = candidates[_hash_to_int([word, str(seed)]) % len(candidates)]
        return word[:pos] + typo + word[pos + 1 :]

    def _corrupt_autocomplete(self, word: str, seed: int) -> str:
        lw = word.lower()
        k = min(MAX_COMMON_START, max(MIN_COMMON_START, len(lw)))
        for kk in range(k, MIN_COMMON_START - 1, -1):
            bag = self.start_bags.get(kk, {}).get(lw[:kk], [])
            candidates = [c for c in bag if c != lw]
            if candidates:
                return candidates[_hash_to_int([word, str(seed)]) % len(candidates)]
        return self._corrupt_autocorrect(word, seed)

    def _corrupt_autocorrect(self, word: str, seed: int) -> str:
        i = self.word_index.get(word.lower())
        if i is None:
            return word  # not a common word: corruption "fails" (reference)
        row = self.distances[i].copy()
        row[i] = 255
        best = int(row.min())
        if best == 255:
            return word
        candidates = np.where(row == best)[0]
        pick = candidates[_hash_to_int([word, str(seed)]) % len(candidates)]
        return self.common_words[int(pick)]

    def _corrupt_synonym(self, word: str, seed: int) -> str:
        syns = self.synonyms.get(word.lower())
        if not syns:
            return self._corrupt_autocorrect(word, seed)
        return syns[_hash_to_int([word, str(seed)]) % len(syns)]
