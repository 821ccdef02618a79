"""Word tokenizer (keras ``Tokenizer`` role for the IMDB case study —
reference case_study_imdb.py:321-336): frequency-ranked vocabulary with an
index budget, lowercasing, punctuation filtering, and fixed-length padded
sequences."""

import collections
import re
from typing import Dict, List

import numpy as np

_FILTER = re.compile(r"[!\"#$%&()*+,\-./:;<=>?@\[\\\]^_`{|}~\t\n]")


def _split(text: str) -> List[str]:
    return _FILTER.sub(" ", text.lower()).split()


class Tokenizer:
    """Frequency-ranked word index (1-based; 0 reserved for padding)."""

    def __init__(self, num_words: int = 2000):
        self.num_words = num_words
        self.word_index: Dict[str, int] = {}

    def fit_on_texts(self, texts: List[str]) -> "Tokenizer":
        counts = collections.Counter(w for t in texts for w in _split(t))
        # rank by frequency (ties by insertion order, like keras)
        for i, (w, _) in enumerate(counts.most_common(), start=1):
            self.word_index[w] = i
        return self

    def texts_to_sequences(self, texts: List[str]) -> List[List[int]]:
        """Word-index sequences; words outside the num_words budget (or
        unseen) are dropped (keras default, no OOV token)."""
        out = []
        for t in texts:
            seq = []
            for w in _split(t):
                i = self.word_index.get(w)
                if i is not None and i < self.num_words:
                    seq.append(i)
            out.append(seq)
        return out


def pad_sequences(seqs: List[List[int]], maxlen: int) -> np.ndarray:
    """Pre-pad/pre-truncate to maxlen (keras pad_sequences defaults)."""
    out = np.zeros((len(seqs), maxlen), dtype=np.int64)
    for i, s in enumerate(seqs):
        s = s[-maxlen:]
        if s:
            out[i, maxlen - len(s):] = np.asarray(s, dtype=np.int64)
    return out
