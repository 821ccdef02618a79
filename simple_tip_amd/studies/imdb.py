"""IMDB case study: 1-block transformer over synthetic token sequences.

Reference parity: case_study_imdb.py (vocab 2000, maxlen 100, 10 epochs,
batch 32, predict batch 600, dsa_badge_size 500, num_selected 2500,
SA [5], NC [3, 5] — the int-valued tap entries; the reference's tuple
entries are dead config)."""

import numpy as np

from ..config import StudyConfig
from ..models.transformer import ImdbTransformer
from .base import CaseStudy
from .synthetic import corrupt_tokens, make_ood_split, synthetic_tokens

VOCAB_SIZE = 2000
INPUT_MAXLEN = 100


class ImdbCaseStudy(CaseStudy):
    def __init__(self, **kw):
        self.config = StudyConfig(
            name="imdb",
            num_classes=2,
            input_shape=(INPUT_MAXLEN,),
            train_size=25000,
            test_size=25000,
            sa_layers=[5],
            nc_layers=[3, 5],
            epochs=10,
            train_batch=32,
            predict_batch=600,
            num_selected=2500,
            dsa_badge_size=500,
        )
        super().__init__(**kw)

    def build_model(self):
        return ImdbTransformer(vocab_size=VOCAB_SIZE, maxlen=INPUT_MAXLEN)

    def load_text_datasets(self, n_train=None, n_test=None, severity=0.5, seed=0):
        """Full reference-shaped text pipeline: synthetic texts ->
        TextCorruptor (IMDB-C, severity 0.5, seed 0 — reference
        case_study_imdb.py:316-319) -> Tokenizer(2000) -> pad to maxlen
        (case_study_imdb.py:321-336). Returns token arrays like
        :meth:`load_datasets`.
        """
        from ..core.text_corruptor import TextCorruptor
        from ..utils.tokenizer import Tokenizer, pad_sequences
        from .synthetic import _rng, make_ood_split

        cfg = self.config
        n_train = n_train or self._n(cfg.train_size)
        n_test = n_test or self._n(cfg.test_size)

        # deterministic pseudo-word vocabulary with class-dependent usage
        vrng = _rng(cfg.name, "text-vocab")
        letters = "abcdefghijklmnopqrstuvwxyz"
        vocab = sorted(
            {
                "".join(vrng.choice(list(letters), size=vrng.randint(5, 9)))
                for _ in range(3000)
            }
        )
        boost = _rng(cfg.name, "text-classes").rand(cfg.num_classes, len(vocab)) ** 4
        base = 1.0 / (np.arange(len(vocab)) + 10.0)

        def make_texts(split, n):
            rng = _rng(cfg.name, split)
            y = rng.randint(0, cfg.num_classes, size=n)
            texts = []
            for i in range(n):
                p = base * (1.0 + 8.0 * boost[y[i]])
                p /= p.sum()
                words = rng.choice(vocab, size=INPUT_MAXLEN + 20, p=p)
                texts.append(" ".join(words))
            return texts, y.astype(np.int64)

        train_texts, train_y = make_texts("text-train", n_train)
        nom_texts, nom_y = make_texts("text-test", n_test)
        cor_src, cor_y = make_texts("text-corrupt-src", n_test)
        corruptor = TextCorruptor(train_texts, dict_size=2000)
        cor_texts = corruptor.corrupt(cor_src, severity=severity, seed=seed)

        tok = Tokenizer(num_words=VOCAB_SIZE).fit_on_texts(train_texts)
        to_arr = lambda ts: pad_sequences(tok.texts_to_sequences(ts), INPUT_MAXLEN)
        train = (to_arr(train_texts), train_y)
        nominal = (to_arr(nom_texts), nom_y)
        ood = make_ood_split(nominal[0], nominal[1], to_arr(cor_texts), cor_y)
        return train, nominal, ood

    def load_datasets(self):
        cfg = self.config
        n_train = self._n(cfg.train_size)
        n_test = self._n(cfg.test_size)
        train = synthetic_tokens(
            cfg.name, "train", n_train, INPUT_MAXLEN, VOCAB_SIZE, cfg.num_classes
        )
        nominal = synthetic_tokens(
            cfg.name, "test", n_test, INPUT_MAXLEN, VOCAB_SIZE, cfg.num_classes
        )
        raw_x, raw_y = synthetic_tokens(
            cfg.name, "corrupt-src", n_test, INPUT_MAXLEN, VOCAB_SIZE, cfg.num_classes
        )
        cor_x = corrupt_tokens(cfg.name, raw_x, VOCAB_SIZE, severity=0.5)
        ood = make_ood_split(nominal[0], nominal[1], cor_x, raw_y)
        return train, nominal, ood
