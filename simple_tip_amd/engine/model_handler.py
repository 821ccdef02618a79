"""Model-centric utilities: activation collection and uncertainty scoring.

Capability parity with reference src/dnn_test_prio/handler_model.py:88-206.
MI355X-native differences:
- activation taps come out of the single forward pass (models/base.py), not a
  second "transparent" keras model; tensors stay on device;
- the four point-prediction quantifiers are one fused epilogue over the
  softmax (ops.softmax_uncertainties — the K14 kernel on device);
- MC-dropout variation ratio runs DROPOUT_SAMPLE_SIZE stochastic forwards
  with device RNG and on-device vote counting.
Timing taxonomy is the reference's: per metric [setup, pred, quant, cam].
"""

import logging
from typing import Dict, Generator, List, Optional, Sequence, Tuple

import numpy as np
import torch
import torch.nn as nn

from ..config import DROPOUT_SAMPLE_SIZE
from ..core.timer import DeviceTimer
from .. import ops
from ..models.base import TapModel

logger = logging.getLogger(__name__)


def iter_batches(x, batch_size: int):
    """Yield contiguous batches of a tensor/ndarray along dim 0."""
    n = x.shape[0]
    for s in range(0, n, batch_size):
        yield x[s : s + batch_size]


class BaseModel:
    """Wraps a :class:`TapModel` for AT extraction and uncertainty scoring."""

    def __init__(
        self,
        model: TapModel,
        activation_layers: Optional[Sequence[int]],
        include_last_layer: bool = False,
        device: Optional[torch.device] = None,
        predict_batch: int = 512,
    ):
        self.model = model
        self.activation_layers = list(activation_layers) if activation_layers else None
        self.include_last_layer = include_last_layer
        self.device = device or next(model.parameters()).device
        self.predict_batch = predict_batch
        self._graphed = None  # lazy GraphedExtractor (GPU fast path, K15)
        self.model.eval()

    def _to_device(self, x) -> torch.Tensor:
        if not isinstance(x, torch.Tensor):
            x = torch.from_numpy(np.ascontiguousarray(x))
        if x.dtype == torch.float64:
            x = x.float()
        return x.to(self.device, non_blocking=True)

    # -- activation extraction (K15) ------------------------------------

    @torch.no_grad()
    def get_activations(self, x) -> List[torch.Tensor]:
        """Single pass over (possibly large) input; returns per-layer AT
        tensors (+ softmax output last iff include_last_layer)."""
        outs: Optional[List[List[torch.Tensor]]] = None
        for batch in self.walk_activations(iter_batches(x, self.predict_batch)):
            if outs is None:
                outs = [[t] for t in batch]
            else:
                for acc, t in zip(outs, batch):
                    acc.append(t)
        assert outs is not None, "empty dataset"
        return [torch.cat(parts, dim=0) for parts in outs]

    @torch.no_grad()
    def walk_activations(self, batches) -> Generator[List[torch.Tensor], None, None]:
        """Stream batches -> per-batch list of tapped activation tensors.

        On GPU this routes through the BN-folded, hipGraph-replayed
        :class:`~simple_tip_amd.engine.extractor.GraphedExtractor` (bf16
        channels_last; fused ResNet-20 block kernels where applicable) —
        the eager MIOpen path was launch-bound.
        ``TIP_NO_GRAPH_EXTRACTOR=1`` restores the eager path for debugging.
        """
        import os

        if self.activation_layers is None:
            raise ValueError("No activation layers specified")
        if (
            self.device.type == "cuda"
            and os.environ.get("TIP_NO_GRAPH_EXTRACTOR") != "1"
        ):
            if self._graphed is None:
                from .extractor import GraphedExtractor

                self._graphed = GraphedExtractor(
                    self.model, self.activation_layers, self.device,
                    self.predict_batch,
                )
            for batch in batches:
                taps, probs = self._graphed(batch)
                out = list(taps)
                if self.include_last_layer:
                    out.append(probs)
                yield out
            return
        for batch in batches:
            xb = self._to_device(batch)
            taps, logits = self.model.forward_taps(xb, self.activation_layers)
            out = list(taps)
            if self.include_last_layer:
                out.append(torch.softmax(logits.float(), dim=1))
            yield out

    # -- uncertainty quantification (K14 + MC-dropout VR) ----------------

    @torch.no_grad()
    def get_pred_and_uncertainty(
        self, x
    ) -> Tuple[np.ndarray, Dict[str, np.ndarray], Dict[str, List[float]]]:
        """Point predictions + all uncertainty scores + per-metric times.

        Returns (pred, {metric: scores}, {metric: [setup, pred, quant, cam]}).
        """
        pred_timer = DeviceTimer()
        quant_timer = DeviceTimer()
        unc_parts: Dict[str, List[torch.Tensor]] = {}
        preds_parts: List[torch.Tensor] = []
        for batch in iter_batches(x, self.predict_batch):
            with pred_timer:
                xb = self._to_device(batch)
                logits = self.model(xb)
                probs = torch.softmax(logits.float(), dim=1)
            with quant_timer:
                scores = ops.softmax_uncertainties(probs)
            preds_parts.append(probs.argmax(dim=1))
            for k, v in scores.items():
                unc_parts.setdefault(k, []).append(v)

        pred = torch.cat(preds_parts).cpu().numpy()
        uncertainties = {
            k: torch.cat(v).float().cpu().numpy() for k, v in unc_parts.items()
        }
        times = {
            k: [0.0, pred_timer.get(), quant_timer.get(), 0.0] for k in uncertainties
        }

        if self.model.has_dropout():
            vr, vr_times = self._variation_ratio(x)
            uncertainties["VR"] = vr
            times["VR"] = vr_times
        else:
            logger.warning(
                "No stochastic (dropout) layers in model; skipping VR."
            )
        return pred, uncertainties, times

    @torch.no_grad()
    def _variation_ratio(self, x) -> Tuple[np.ndarray, List[float]]:
        """MC-dropout variation ratio over DROPOUT_SAMPLE_SIZE forwards."""
        sampling_timer = DeviceTimer()
        quant_timer = DeviceTimer()
        self.model.eval()
        # enable ONLY dropout stochasticity
        for m in self.model.modules():
            if isinstance(m, nn.Dropout):
                m.train()
        try:
            n = x.shape[0]
            num_classes = self.model.num_classes
            counts = torch.zeros(
                n, num_classes, dtype=torch.float32, device=self.device
            )
            with sampling_timer:
                for s0 in range(0, n, self.predict_batch):
                    xb = self._to_device(x[s0 : s0 + self.predict_batch])
                    b = xb.shape[0]
                    votes = torch.zeros(
                        b, num_classes, dtype=torch.float32, device=self.device
                    )
                    # vectorize the MC samples: R independent dropout
                    # replicas per forward pass (throughput, not 200 passes)
                    # macro-batch cap 8192: measured optimum (32768 made
                    # MIOpen pick a 5x slower conv algo for these shapes)
                    reps = max(1, min(DROPOUT_SAMPLE_SIZE, 8192 // max(b, 1)))
                    done = 0
                    use_amp = self.device.type == "cuda"
                    while done < DROPOUT_SAMPLE_SIZE:
                        r = min(reps, DROPOUT_SAMPLE_SIZE - done)
                        xrep = xb.repeat(r, *([1] * (xb.dim() - 1)))
                        # bf16 autocast: the VR statistic is an argmax vote,
                        # robust to reduced-precision logits
                        with torch.autocast("cuda", dtype=torch.bfloat16,
                                            enabled=use_amp):
                            logits = self.model(xrep)
                        preds = logits.argmax(dim=1).reshape(r, b)
                        votes.scatter_add_(
                            1,
                            preds.t(),
                            torch.ones(b, r, device=self.device),
                        )
                        done += r
                    counts[s0 : s0 + b] = votes
            with quant_timer:
                top = counts.max(dim=1).values
                vr = 1.0 - top / float(DROPOUT_SAMPLE_SIZE)
            return (
                vr.cpu().numpy(),
                [0.0, sampling_timer.get(), quant_timer.get(), 0.0],
            )
        finally:
            self.model.eval()
