"""Autograd wrapper for the fused focal-L2 loss kernels."""
from __future__ import annotations

import torch

from ._backend import hip_extension

_NW_CACHE = {}


def _nstack_weight_tensor(nstack_weight, device):
    """Constant per config — cached on device (a per-call as_tensor would do a
    pageable H2D copy, which hipGraph capture forbids)."""
    key = (tuple(nstack_weight), device)
    t = _NW_CACHE.get(key)
    if t is None:
        t = torch.as_tensor(list(nstack_weight), dtype=torch.float32,
                            device=device)
        _NW_CACHE[key] = t
    return t


class FocalL2Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, pred, gt, mask, heat_start, bkg_start, gamma,
                mtw, ktw, nstack_weight, alpha, beta):
        ext = hip_extension()
        pred = pred.contiguous()
        gt = gt.contiguous()
        mask = mask.contiguous()
        sums = ext.focal_l2_fwd(pred, gt, mask, heat_start, bkg_start, gamma,
                                mtw, ktw, alpha, beta)
        nw = _nstack_weight_tensor(nstack_weight, pred.device)
        loss = (sums * nw).sum() / nw.sum()
        ctx.save_for_backward(pred, gt, mask, nw)
        ctx.conf = (heat_start, bkg_start, gamma, mtw, ktw, alpha, beta)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        ext = hip_extension()
        pred, gt, mask, nw = ctx.saved_tensors
        heat_start, bkg_start, gamma, mtw, ktw, alpha, beta = ctx.conf
        gscale = (dloss.detach().float() * nw / nw.sum()).contiguous()
        dpred = ext.focal_l2_bwd(pred, gt, mask, gscale, heat_start, bkg_start,
                                 gamma, mtw, ktw, alpha, beta)
        return (dpred, None, None, None, None, None, None, None, None, None, None)


def focal_l2_loss_hip(pred, gt, mask, *, heat_start, bkg_start, gamma=1,
                      multi_task_weight=0.1, keypoint_task_weight=3.0,
                      nstack_weight=(1, 1, 1, 1), alpha=0.0, beta=0.0):
    return FocalL2Fn.apply(pred, gt, mask, heat_start, bkg_start, gamma,
                           multi_task_weight, keypoint_task_weight,
                           tuple(nstack_weight), alpha, beta)
