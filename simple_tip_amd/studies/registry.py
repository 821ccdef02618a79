"""Case-study registry keyed by the reference's CLI names
(reproduction.py CaseStudyType values, plus the benchmark flagship)."""

from .image_studies import (
    Cifar10CaseStudy,
    Cifar10ResNetCaseStudy,
    FashionMnistCaseStudy,
    MnistCaseStudy,
)
from .imdb import ImdbCaseStudy

STUDIES = {
    "mnist": MnistCaseStudy,
    "fmnist": FashionMnistCaseStudy,
    "cifar10": Cifar10CaseStudy,
    "cifar10_resnet": Cifar10ResNetCaseStudy,
    "imdb": ImdbCaseStudy,
}


def get_case_study(name: str, **kwargs):
    """Instantiate a case study by CLI name."""
    try:
        cls = STUDIES[name]
    except KeyError:
        raise ValueError(f"Unknown case study: {name} (have {sorted(STUDIES)})")
    return cls(**kwargs)
