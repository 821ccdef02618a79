"""The ops dispatcher's loud-failure contract: a GPU tensor with no HIP
extension must RAISE (no silent eager fallback on a GPU box)."""

import numpy as np
import pytest
import torch

import simple_tip_amd.ops as ops


class _FakeCudaTensor:
    is_cuda = True


def test_gpu_without_extension_raises(monkeypatch):
    monkeypatch.setattr(ops, "_load_ext", lambda: None)
    monkeypatch.delenv("TIP_ALLOW_GPU_FALLBACK", raising=False)
    with pytest.raises(RuntimeError, match="HIP extension"):
        ops._route(_FakeCudaTensor())


def test_gpu_fallback_only_with_explicit_optin(monkeypatch):
    monkeypatch.setattr(ops, "_load_ext", lambda: None)
    monkeypatch.setenv("TIP_ALLOW_GPU_FALLBACK", "1")
    assert ops._route(_FakeCudaTensor()) is ops.fallback


def test_cpu_routes_to_fallback():
    assert ops._route(torch.zeros(2)) is ops.fallback


def test_cam_order_no_bits():
    # profiles with zero coverable bits: pure score ordering
    words = torch.zeros(4, 1, dtype=torch.int64)
    scores = torch.tensor([0.1, 0.9, 0.5, 0.7])
    order = ops.cam_order(scores, words, nbits=3)
    assert order.tolist() == [1, 3, 2, 0]


def test_bench_requires_extension_message():
    # the bench guards explicitly (documented contract)
    src = open("bench.py").read()
    assert "requires the _tip_hip extension" in src
