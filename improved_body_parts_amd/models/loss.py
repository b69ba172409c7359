"""Multi-task focal-L2 loss over 4 stacks x 5 scales.

Capability parity with reference models/loss_model.py (MultiTaskLoss :7-161) and
models/loss_model_parallel.py (MultiTaskLossParallel). Semantics preserved:

  * GT heatmaps are down-scaled to each prediction scale with adaptive average
    pooling; mask_miss is bilinearly resized then thresholded at 0.5
    (reference loss_model.py:52-56).
  * focal L2: ``st = where(gt >= 0.01, s - alpha, 1 - s - beta)``,
    ``factor = |1 - st| ** gamma`` with gamma=1 by default (the reference DDP
    path, loss_model.py:151-152); gamma=2 reproduces the paper/README headline
    variant (loss_model_parallel.py:89-90). Exposed as a config knob as SURVEY
    §7.3 prescribes.
  * mask channel weighting: person-mask channel x multi_task_weight, keypoint
    channels x keypoint_task_weight (loss_model.py:146-149).
  * weighted sum over scales (scale_weight) and stacks (nstack_weight), divided
    by batch size (loss_model.py:34-40).

Re-designed for MI355X: the whole per-scale loss (mask broadcast + task weights
+ focal factor + reduction, fwd AND bwd) runs as one fused HIP kernel via
``ops.focal_l2_loss`` — the reference materialises several (nstack,N,C,H,W)
temporaries instead.
"""
from __future__ import annotations

import torch
import torch.nn.functional as F
from torch import nn

from .. import ops


class MultiTaskLoss(nn.Module):
    """Per-process loss used in distributed training (reference loss_model.py)."""

    def __init__(self, opt, config, heatmap_weight=1, offset_weight=1, gamma=1, **kwargs):
        super().__init__()
        self.nstack = opt.nstack
        self.batch_size = opt.batch_size
        self.offset_start = config.offset_start
        self.heat_start = config.heat_start
        self.bkg_start = config.bkg_start
        self.multi_task_weight = opt.multi_task_weight
        self.keypoint_task_weight = opt.keypoint_task_weight
        self.scale_weight = opt.scale_weight
        self.nstack_weight = opt.nstack_weight
        self.heatmap_weight = heatmap_weight
        self.offset_weight = offset_weight
        self.gamma = gamma

    def forward(self, pred_tuple, target_tuple):
        """pred_tuple: [nstack][n_scales] tensors (N,C,Hs,Ws); target_tuple:
        (mask_miss (N,1,H,W), heatmaps (N,C,H,W)). Returns a scalar loss
        (averaged over batch). n_scales is 5 for the IMHN families and 1 for
        the single-scale AE variant."""
        nstack = len(pred_tuple)
        n_scales = len(pred_tuple[0])
        batch = pred_tuple[0][0].shape[0]
        sw = self.scale_weight[:n_scales] if n_scales > 1 else [1.0]
        loss_scales = []
        for i in range(n_scales):
            pred = torch.stack([pred_tuple[j][i] for j in range(nstack)], dim=0)
            loss_scales.append(self._loss_per_scale(pred, target_tuple) * sw[i])
        return sum(loss_scales) / sum(sw) / batch

    def _loss_per_scale(self, pred, target):
        # gt/mask go in at FULL resolution: the fused kernel average-pools GT
        # windows and bilinearly samples + thresholds mask_miss on the fly
        # (reference loss_model.py:52-56 materialises a pyramid per scale;
        # the eager fallback inside ops.focal_l2_loss reproduces it exactly)
        return ops.focal_l2_loss(
            pred, target[1].to(pred.dtype), target[0].to(pred.dtype),
            heat_start=self.heat_start, bkg_start=self.bkg_start,
            gamma=self.gamma,
            multi_task_weight=self.multi_task_weight,
            keypoint_task_weight=self.keypoint_task_weight,
            nstack_weight=self.nstack_weight)

    # -- reference-parity plain losses (loss_model.py:83-131) -------------------
    @staticmethod
    def l2_loss(s, sxing, mask_miss, heat_start, bkg_start, multi_task_weight=0.1,
                keypoint_task_weight=1, nstack_weight=(1, 1, 1, 1)):
        mask = mask_miss.expand_as(sxing).clone()
        mask[:, :, -2, :, :] = mask[:, :, -2, :, :] * multi_task_weight
        mask[:, :, heat_start:bkg_start, :, :] = \
            mask[:, :, heat_start:bkg_start, :, :] * keypoint_task_weight
        out = (s - sxing) ** 2 * mask
        loss_nstack = out.sum(dim=(1, 2, 3, 4))
        w = [loss_nstack[i] * nstack_weight[i] for i in range(len(nstack_weight))]
        return sum(w) / sum(nstack_weight)

    @staticmethod
    def l1_loss(pred, target, mask_offset, nstack_weight=(1, 1, 1, 1)):
        out = torch.abs(pred - target) * mask_offset
        loss_nstack = out.sum(dim=(1, 2, 3, 4))
        w = [loss_nstack[i] * nstack_weight[i] for i in range(len(nstack_weight))]
        return sum(w) / sum(nstack_weight)


class MultiTaskLossParallel(MultiTaskLoss):
    """Single-process / DataParallel-path loss (reference loss_model_parallel.py).

    Semantics differ from the DDP loss and are preserved exactly (verified
    against the reference in tests/test_reference_parity.py):
      * default is PLAIN L2 (reference :68), focal (gamma=2, no alpha/beta,
        no task weights) available via ``use_focal=True`` (:77-99);
      * mask_miss is bilinear-interpolated WITHOUT the 0.5 threshold and
        broadcast over channels — no per-channel task weighting;
      * no batch division (the driver divides, reference train_parallel.py:145).
    """

    def __init__(self, opt, config, use_focal=False, gamma=2, **kwargs):
        kwargs.setdefault("gamma", gamma)
        super().__init__(opt, config, **kwargs)
        self.use_focal = use_focal

    def forward(self, pred_tuple, target_tuple):
        nstack = len(pred_tuple)
        n_scales = len(pred_tuple[0])
        sw = self.scale_weight[:n_scales] if n_scales > 1 else [1.0]
        loss_scales = []
        for i in range(n_scales):
            pred = torch.stack([pred_tuple[j][i] for j in range(nstack)], dim=0)
            loss_scales.append(self._loss_per_scale(pred, target_tuple) * sw[i])
        return sum(loss_scales) / sum(sw)

    def _loss_per_scale(self, pred, target):
        size = pred.shape[-2:]
        mask = F.interpolate(target[0].float(), size=size, mode="bilinear",
                             align_corners=False)
        gt = F.adaptive_avg_pool2d(target[1].float(), output_size=size)
        pred = pred.float()
        if self.use_focal:
            st = torch.where(gt[None] >= 0.01, pred, 1 - pred)
            factor = (1.0 - st) ** self.gamma
            out = (pred - gt[None]) ** 2 * factor * mask[None]
        else:
            out = (pred - gt[None]) ** 2 * mask[None]
        loss_nstack = out.sum(dim=(1, 2, 3, 4))
        w = [loss_nstack[i] * self.nstack_weight[i]
             for i in range(len(self.nstack_weight))]
        return sum(w) / sum(self.nstack_weight)
