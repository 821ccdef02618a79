"""Neuron-coverage orchestration per model.

Capability parity with reference src/dnn_test_prio/handler_coverage.py:20-205
(same 12 configured metrics with the same time-debit accounting). MI355X
re-design: profiles are packed bitmaps held resident in HBM (288 GB) instead
of bool .npy spills to /assets/.tmp (reference handler_coverage.py:165-205),
and the CAM greedy loop runs on the device bitmap (ops.cam_order).
"""

import logging
from typing import Callable, Dict, List, Tuple

import numpy as np
import torch

from ..core.bitmap import BitProfile
from ..core.neuron_coverage import KMNC, NAC, NBC, SNAC, TKNC, CoverageMethod
from ..core.prioritizers import cam
from ..core.timer import DeviceTimer
from .model_handler import BaseModel, iter_batches

logger = logging.getLogger(__name__)


class CoverageWorker:
    """Fits aggregate train statistics once; evaluates all NC metrics.

    ``dist_shard=True`` on an initialised torch.distributed group shards
    BOTH passes across ranks: the train-statistics pass walks this rank's
    row shard and Chan/min/max-merges the partials (K18 collectives), and
    ``evaluate_all`` profiles this rank's test shard, reassembling the full
    profile matrix with the coverage-bitmap OR all-reduce (BASELINE
    config 4) so scores and the CAM order are identical on every rank.
    """

    def __init__(self, base_model: BaseModel, training_set, dist_shard: bool = False):
        from ..parallel.dist import get_world_size, shard_slice
        from .aggregate_statistics import AggregateStatisticsCollector

        self.base_model = base_model
        self.dist_shard = bool(dist_shard) and get_world_size() > 1
        self.metrics: Dict[str, CoverageMethod] = {}
        self.setup_times: Dict[str, float] = {}

        train_part = (
            training_set[shard_slice(training_set.shape[0])]
            if self.dist_shard
            else training_set
        )
        agg = AggregateStatisticsCollector()
        pred_timer = DeviceTimer(start=True)
        for acts in base_model.walk_activations(
            iter_batches(train_part, base_model.predict_batch)
        ):
            pred_timer.stop()
            agg.track(acts)
            pred_timer.start()
        pred_timer.stop()
        if self.dist_shard:
            agg.cross_rank_merge()
        mins, maxs, stds = agg.get()

        nbc_debit = (
            agg.min_timer.get()
            + agg.max_timer.get()
            + pred_timer.get()
            + agg.welford_timer.get()
        )
        for scaler in (0, 0.5, 1):
            self._add_metric(
                f"NBC_{scaler:g}",
                lambda scaler=scaler: NBC(mins=mins, maxs=maxs, stds=stds, scaler=scaler),
                time_debit=nbc_debit,
            )
        snac_debit = (
            agg.welford_timer.get() + agg.max_timer.get() + pred_timer.get()
        )
        for scaler in (0, 0.5, 1):
            self._add_metric(
                f"SNAC_{scaler:g}",
                lambda scaler=scaler: SNAC(maxs=maxs, stds=stds, scaler=scaler),
                time_debit=snac_debit,
            )
        self._add_metric("NAC_0", lambda: NAC(cov_threshold=0.0))
        self._add_metric("NAC_0.75", lambda: NAC(cov_threshold=0.75))
        for k in (1, 2, 3):
            self._add_metric(f"TKNC_{k}", lambda k=k: TKNC(top_neurons=k))
        kmnc_debit = agg.min_timer.get() + agg.max_timer.get() + pred_timer.get()
        self._add_metric(
            "KMNC_2", lambda: KMNC(mins, maxs, sections=2), time_debit=kmnc_debit
        )

    def _add_metric(
        self,
        metric_id: str,
        metric_supplier: Callable[[], CoverageMethod],
        time_debit: float = 0.0,
    ):
        timer = DeviceTimer()
        with timer:
            self.metrics[metric_id] = metric_supplier()
        self.setup_times[metric_id] = time_debit + timer.get()

    def evaluate_all(
        self, test_dataset, test_dataset_id: str
    ) -> Tuple[Dict[str, List[float]], Dict[str, np.ndarray], Dict[str, List[int]]]:
        """Per metric: [setup, pred, quant(, cam)] times, scores, CAM orders."""
        times: Dict[str, List[float]] = {
            m: [t, 0.0, 0.0] for m, t in self.setup_times.items()
        }
        scores_parts: Dict[str, List[torch.Tensor]] = {m: [] for m in self.metrics}
        profile_parts: Dict[str, List[BitProfile]] = {m: [] for m in self.metrics}

        n_total = test_dataset.shape[0]
        if self.dist_shard:
            from ..parallel.dist import shard_slice

            my_slice = shard_slice(n_total)
            test_part = test_dataset[my_slice]
        else:
            test_part = test_dataset
        gen = self.base_model.walk_activations(
            iter_batches(test_part, self.base_model.predict_batch)
        )
        while True:
            t = DeviceTimer()
            try:
                with t:
                    acts = next(gen)
            except StopIteration:
                break
            pred_time = t.get()
            for metric_id, metric in self.metrics.items():
                qt = DeviceTimer()
                with qt:
                    s, p = metric(acts)
                times[metric_id][1] += pred_time
                times[metric_id][2] += qt.get()
                scores_parts[metric_id].append(s)
                profile_parts[metric_id].append(p)

        all_scores: Dict[str, np.ndarray] = {}
        cam_orders: Dict[str, List[int]] = {}
        for metric_id in self.metrics.keys():
            scores = torch.cat(scores_parts[metric_id])
            profile = BitProfile.cat(profile_parts[metric_id])
            if self.dist_shard:
                scores, profile = self._assemble_shards(
                    scores, profile, n_total, my_slice
                )
            all_scores[metric_id] = scores.cpu().numpy()
            logger.info("Calculating CAM for %s (%s)", metric_id, test_dataset_id)
            timer = DeviceTimer()
            with timer:
                order = list(cam(scores.float(), profile))
            times[metric_id].append(timer.get())
            self._cam_sanity_check(order, all_scores[metric_id])
            cam_orders[metric_id] = order
        return times, all_scores, cam_orders

    @staticmethod
    def _assemble_shards(scores, profile: BitProfile, n_total: int, my_slice):
        """Reassemble full (scores, profiles) from per-rank input shards.

        The profile rows land in a zero-initialised full-size word matrix and
        a bitwise-OR all-reduce unions them — rows are disjoint across ranks,
        so the OR is exact and every rank ends with the identical full
        matrix for CAM. Scores all-gather (reference semantics
        handler_coverage.py:189-205, sharded per SURVEY §2.4).
        """
        from ..parallel.dist import allgather_rows
        from ..parallel.sharded import allreduce_bitmap_or

        full_words = torch.zeros(
            n_total, profile.words.shape[1],
            dtype=profile.words.dtype, device=profile.words.device,
        )
        full_words[my_slice] = profile.words
        allreduce_bitmap_or(full_words)
        full_scores = allgather_rows(scores, n_total)
        return full_scores, BitProfile(full_words, profile.nbits)

    @staticmethod
    def _cam_sanity_check(cam_order, scores):
        assert (
            len(cam_order) == len(set(cam_order)) == scores.shape[0]
        ), "CAM order is not unique or not complete"
