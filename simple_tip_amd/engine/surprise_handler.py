"""Surprise-adequacy orchestration per model.

Capability parity with reference src/dnn_test_prio/handler_surprise.py:19-117
(same TESTED_SA configs, timing taxonomy [setup, pred, quant, cam], dynamic
surprise-coverage upper bound, per-dataset CAM). Train/test ATs stay resident
on device; SA hot loops route through the MFMA pairwise kernel via ops.
"""

import logging
from typing import Dict, List, Optional

import numpy as np
import torch

from ..config import NUM_SC_BUCKETS
from ..core.prioritizers import cam
from ..core.surprise import DSA, LSA, MDSA, MLSA, MultiModalSA, SurpriseCoverageMapper
from ..core.timer import DeviceTimer
from .model_handler import BaseModel

logger = logging.getLogger(__name__)


class SurpriseHandler:
    """Fits on train ATs once, evaluates all SA variants on test sets."""

    TESTED_SA = {
        "dsa": lambda x, y: DSA(x, y, subsampling=0.3),
        "pc-lsa": lambda x, y: MultiModalSA.build_by_class(x, y, lambda a, p: LSA(a)),
        "pc-mdsa": lambda x, y: MultiModalSA.build_by_class(x, y, lambda a, p: MDSA(a)),
        "pc-mlsa": lambda x, y: MultiModalSA.build_by_class(
            x, y, lambda a, p: MLSA(a, num_components=3)
        ),
        "pc-mmdsa": lambda x, y: MultiModalSA.build_with_kmeans(
            x, y, lambda a, p: MDSA(a), potential_k=range(2, 6), subsampling=0.3
        ),
    }

    def __init__(
        self,
        model,
        sa_layers: List[int],
        training_dataset,
        device=None,
        predict_batch: int = 512,
        dist_shard: bool = False,
    ):
        from ..parallel.dist import get_world_size

        self.sa_layers = list(sa_layers)
        # dist_shard: shard AT-extraction and SA scoring over the test-input
        # axis across ranks (DP per SURVEY §2.4); train ATs are extracted
        # sharded then all-gathered so every rank fits identical SAs.
        self.dist_shard = bool(dist_shard) and get_world_size() > 1
        self.base_model = BaseModel(
            model,
            self.sa_layers,
            include_last_layer=True,
            device=device,
            predict_batch=predict_batch,
        )
        self.train_at_timer = DeviceTimer()
        with self.train_at_timer:
            if self.dist_shard:
                from ..parallel.dist import allgather_rows, shard_slice

                n = training_dataset.shape[0]
                ats_l, pred_l = self._acti_and_pred(
                    training_dataset[shard_slice(n)]
                )
                self.train_ats = allgather_rows(ats_l, n)
                self.train_pred = allgather_rows(pred_l, n)
            else:
                self.train_ats, self.train_pred = self._acti_and_pred(
                    training_dataset
                )

    def _acti_and_pred(self, dataset):
        """ATs and argmax predictions in one fused forward pass (K15)."""
        outputs = self.base_model.get_activations(dataset)
        assert len(outputs) == len(self.sa_layers) + 1
        ats = [o.reshape(o.shape[0], -1) for o in outputs[:-1]]
        flat = torch.cat(ats, dim=1) if len(ats) > 1 else ats[0]
        return flat, outputs[-1].argmax(dim=1)

    def evaluate_all(
        self, datasets: Dict[str, object], dsa_badge_size: Optional[int] = None
    ):
        """{sa_name: {ds_name: (scores, cam_order, [setup,pred,quant,cam])}}"""
        res = {}
        test_apt = {}
        for ds_name, dataset in datasets.items():
            t = DeviceTimer()
            with t:
                if self.dist_shard:
                    from ..parallel.dist import shard_slice

                    n = dataset.shape[0]
                    test_ats, test_pred = self._acti_and_pred(
                        dataset[shard_slice(n)]
                    )
                else:
                    n = None
                    test_ats, test_pred = self._acti_and_pred(dataset)
            test_apt[ds_name] = (test_ats, test_pred, t.get(), n)

        for sa_name, sa_func in self.TESTED_SA.items():
            res[sa_name] = {}
            setup_timer = DeviceTimer()
            with setup_timer:
                logger.info("Creating %s instance", sa_name)
                sa = sa_func(self.train_ats, self.train_pred)
                if isinstance(sa, DSA) and dsa_badge_size is not None:
                    sa.badge_size = dsa_badge_size
            setup_time = self.train_at_timer.get() + setup_timer.get()

            for ds_name, (test_ats, test_pred, pred_time, n) in test_apt.items():
                sa_timer = DeviceTimer()
                with sa_timer:
                    logger.info("Calculating %s for %s", sa_name, ds_name)
                    sa_vals = sa(test_ats, test_pred)
                    if self.dist_shard:
                        # publish the per-input score shard (tiny all-gather)
                        from ..parallel.dist import allgather_rows

                        sa_vals = allgather_rows(sa_vals.contiguous(), n)
                res[sa_name][ds_name] = (sa_vals, [setup_time, pred_time, sa_timer.get()])

        for sa_name in self.TESTED_SA.keys():
            for ds_name in datasets.keys():
                sa_vals, times = res[sa_name][ds_name]
                cam_timer = DeviceTimer()
                with cam_timer:
                    vals_t = torch.as_tensor(sa_vals)
                    finite = vals_t[torch.isfinite(vals_t)]
                    upper = float(finite.max()) if finite.numel() else 1.0
                    mapper = SurpriseCoverageMapper(NUM_SC_BUCKETS, upper)
                    profiles = mapper.get_coverage_profile(vals_t)
                    cam_order = np.array(list(cam(vals_t.float(), profiles)))
                times = times + [cam_timer.get()]
                sa_np = vals_t.double().cpu().numpy()
                res[sa_name][ds_name] = (sa_np, cam_order, times)
        return res
