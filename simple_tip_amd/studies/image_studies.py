"""MNIST / Fashion-MNIST / CIFAR-10 case studies (synthetic data) plus the
ResNet-20 benchmark flagship.

Shapes, split sizes, tap layers and hyper-parameters match the reference
(case_study_mnist.py:25-29,50-69; case_study_fashion_mnist.py:29-48;
case_study_cifar10.py:24-26,33-57; dataset sizes per BASELINE.md)."""

from typing import Tuple

import numpy as np

from ..config import StudyConfig
from ..models.cnn import Cifar10CNN, MnistCNN, ResNet20
from .base import CaseStudy
from .synthetic import corrupt_images, make_ood_split, synthetic_images


class _ImageStudy(CaseStudy):
    model_cls = MnistCNN

    def build_model(self):
        return self.model_cls()

    def load_datasets(self):
        cfg = self.config
        name = cfg.name
        n_train = self._n(cfg.train_size)
        n_test = self._n(cfg.test_size)
        shape = cfg.input_shape
        train = synthetic_images(name, "train", n_train, shape, cfg.num_classes)
        nominal = synthetic_images(name, "test", n_test, shape, cfg.num_classes)
        # corrupted split: same size as nominal, then the reference OOD
        # recipe (corrupted ++ nominal, shuffled seed 0 -> 2x test size)
        raw_x, raw_y = synthetic_images(name, "corrupt-src", n_test, shape, cfg.num_classes)
        cor_x = corrupt_images(name, raw_x, severity=0.5)
        ood = make_ood_split(nominal[0], nominal[1], cor_x, raw_y)
        return train, nominal, ood


class MnistCaseStudy(_ImageStudy):
    """MNIST: 4-layer CNN, 15 epochs, batch 128; SA [3], NC [0..3]."""

    model_cls = MnistCNN

    def __init__(self, **kw):
        self.config = StudyConfig(
            name="mnist",
            num_classes=10,
            input_shape=(1, 28, 28),
            train_size=60000,
            test_size=10000,
            sa_layers=[3],
            nc_layers=[0, 1, 2, 3],
            epochs=15,
            train_batch=128,
            num_selected=1000,
        )
        super().__init__(**kw)


class FashionMnistCaseStudy(_ImageStudy):
    """Fashion-MNIST: same CNN/hparams as MNIST (reference
    case_study_fashion_mnist.py:29-48)."""

    model_cls = MnistCNN

    def __init__(self, **kw):
        self.config = StudyConfig(
            name="fmnist",
            num_classes=10,
            input_shape=(1, 28, 28),
            train_size=60000,
            test_size=10000,
            sa_layers=[3],
            nc_layers=[0, 1, 2, 3],
            epochs=15,
            train_batch=128,
            num_selected=1000,
        )
        super().__init__(**kw)


class Cifar10CaseStudy(_ImageStudy):
    """CIFAR-10: TF-tutorial CNN, 20 epochs, batch 32; no dropout => no VR."""

    model_cls = Cifar10CNN

    def __init__(self, **kw):
        self.config = StudyConfig(
            name="cifar10",
            num_classes=10,
            input_shape=(3, 32, 32),
            train_size=50000,
            test_size=10000,
            sa_layers=[3],
            nc_layers=[0, 1, 2, 3],
            epochs=20,
            train_batch=32,
            num_selected=1000,
        )
        super().__init__(**kw)


class Cifar10ResNetCaseStudy(_ImageStudy):
    """CIFAR-10 ResNet-20 — the benchmark flagship (BASELINE.json metric:
    'inputs/sec prioritized (AT+LSA/DSA/Gini) + APFD, CIFAR-10 ResNet')."""

    model_cls = ResNet20

    def __init__(self, **kw):
        self.config = StudyConfig(
            name="cifar10_resnet",
            num_classes=10,
            input_shape=(3, 32, 32),
            train_size=50000,
            test_size=10000,
            sa_layers=list(ResNet20.sa_layers),
            nc_layers=list(ResNet20.nc_layers),
            epochs=20,
            train_batch=128,
            num_selected=1000,
        )
        super().__init__(**kw)
