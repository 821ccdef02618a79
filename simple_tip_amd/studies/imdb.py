"""IMDB case study: 1-block transformer over synthetic token sequences.

Reference parity: case_study_imdb.py (vocab 2000, maxlen 100, 10 epochs,
batch 32, predict batch 600, dsa_badge_size 500, num_selected 2500,
SA [5], NC [3, 5] — the int-valued tap entries; the reference's tuple
entries are dead config)."""

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
