"""Operator layer: every hot op of the framework goes through here.

Two backends:

  * ``hip``   — hand-written CDNA4 (gfx950) kernels in ``csrc/`` compiled into the
    in-tree extension ``_ibp_hip``: MFMA implicit-GEMM convolution (fwd/dgrad/wgrad,
    NHWC bf16), fused BN+LeakyReLU, fused pool/upsample, fused SE, fused focal-L2
    loss fwd+bwd, fused multi-tensor SGD, on-device heatmap GT generation and
    keypoint post-processing. This is THE compute path on MI355X.
  * ``eager`` — plain PyTorch composition. Used on CPU (tests, plumbing config)
    and as the fp32 numerics oracle each kernel is validated against.

Policy: on a CUDA (ROCm) device the HIP extension is REQUIRED — a missing
extension raises rather than silently falling back (set ``IBP_AMD_ALLOW_EAGER=1``
to override for debugging). This replaces the reference's reliance on
cuDNN/Apex with first-party kernels.
"""
from __future__ import annotations

import os

import torch
import torch.nn.functional as F

from . import _backend
from ._backend import (
    hip_available,
    hip_extension,
    require_hip,
    use_hip_for,
)

LEAKY_SLOPE = 0.01


# ---------------------------------------------------------------------------
# conv + bn + activation
# ---------------------------------------------------------------------------

def conv_bn_act(x, conv, bn, act: bool, training: bool,
                residual_post=None, residual_post2=None):
    """conv -> (bn) -> (leaky_relu) (+ residual_post [+ residual_post2]).
    ``conv``/``bn`` are the parameter-holding nn.Conv2d / nn.BatchNorm2d
    modules (or bn None). The post-act residuals ride the conv epilogue on
    the HIP inference path (hourglass skip join, feature-cache add)."""
    if use_hip_for(x):
        from . import conv as _conv
        return _conv.conv_bn_act_hip(x, conv, bn, act=act, residual=None,
                                     training=training,
                                     residual_post=residual_post,
                                     residual_post2=residual_post2)
    y = conv(x)
    if bn is not None:
        y = bn(y)
    if act:
        y = F.leaky_relu(y, LEAKY_SLOPE, inplace=True)
    if residual_post is not None:
        y = y + residual_post
    if residual_post2 is not None:
        y = y + residual_post2
    return y


def conv_bn_add_act(x, conv, bn, residual, act: bool, training: bool,
                    residual_post=None):
    """conv -> (bn) -> += residual -> (leaky_relu); the fused tail of a Residual block."""
    if use_hip_for(x):
        from . import conv as _conv
        return _conv.conv_bn_act_hip(x, conv, bn, act=act, residual=residual,
                                     training=training,
                                     residual_post=residual_post)
    y = conv(x)
    if bn is not None:
        y = bn(y)
    y = y + residual
    if act:
        y = F.leaky_relu(y, LEAKY_SLOPE, inplace=True)
    if residual_post is not None:
        y = y + residual_post
    return y


# ---------------------------------------------------------------------------
# pooling / upsampling / concat
# ---------------------------------------------------------------------------

def maxpool2x2(x):
    if use_hip_for(x):
        from . import spatial
        return spatial.maxpool2x2_hip(x)
    return F.max_pool2d(x, 2, 2)


def upsample2x_nearest(x):
    if use_hip_for(x):
        from . import spatial
        return spatial.upsample2x_hip(x)
    return F.interpolate(x, scale_factor=2, mode="nearest")


def channel_concat(a, b):
    if a.is_cuda:
        return torch.cat([a, b], dim=1).contiguous(memory_format=torch.channels_last)
    return torch.cat([a, b], dim=1)


# ---------------------------------------------------------------------------
# squeeze-excitation
# ---------------------------------------------------------------------------

def se_layer(x, fc1, fc2):
    """Global-avg-pool -> fc1 -> leaky_relu -> fc2 -> sigmoid -> channel scale."""
    if use_hip_for(x):
        from . import spatial
        return spatial.se_layer_hip(x, fc1, fc2)
    n, c = x.shape[:2]
    y = x.float().mean(dim=(2, 3))
    y = F.leaky_relu(fc1(y.to(x.dtype)), LEAKY_SLOPE)
    y = torch.sigmoid(fc2(y))
    return x * y.view(n, c, 1, 1)


# ---------------------------------------------------------------------------
# focal L2 loss (fused fwd/bwd on HIP)
# ---------------------------------------------------------------------------

def focal_l2_loss(pred, gt, mask, *, heat_start, bkg_start, gamma=1,
                  multi_task_weight=0.1, keypoint_task_weight=3.0,
                  nstack_weight=(1, 1, 1, 1), alpha=0.0, beta=0.0):
    """Focal L2 over (nstack, N, C, H, W) predictions.

    ``gt`` (N, C, H0, W0) and ``mask`` (N, 1, H0, W0) may be at an integer
    multiple of the prediction resolution: the HIP kernel average-pools GT
    windows and bilinearly samples mask_miss ON THE FLY (the supervision
    pyramid never materialises — reference loss_model.py:52-56 builds it
    per scale with adaptive_avg_pool2d / interpolate). Mask semantics
    everywhere: bilinear value kept where >= 0.5, else 0. Per-channel task
    weights: person-mask channel (C-2) gets multi_task_weight, keypoint
    channels [heat_start, bkg_start) get keypoint_task_weight
    (reference loss_model.py:146-156).
    """
    if use_hip_for(pred):
        from . import loss as _loss
        h0, w0 = gt.shape[-2:]
        h, w = pred.shape[-2:]
        if h0 % h or w0 % w or h0 // h != w0 // w:
            # non-integer pyramid ratio (odd input sizes): pre-pool eagerly,
            # kernel runs with r = 1
            gt = F.adaptive_avg_pool2d(gt.float(), (h, w)).to(pred.dtype)
            m = F.interpolate(mask.float(), size=(h, w), mode="bilinear",
                              align_corners=False)
            mask = ((m >= 0.5).float() * m).to(pred.dtype)
        return _loss.focal_l2_loss_hip(pred, gt, mask, heat_start=heat_start,
                                       bkg_start=bkg_start, gamma=gamma,
                                       multi_task_weight=multi_task_weight,
                                       keypoint_task_weight=keypoint_task_weight,
                                       nstack_weight=nstack_weight,
                                       alpha=alpha, beta=beta)
    nstack = pred.shape[0]
    C = pred.shape[2]
    size = pred.shape[-2:]
    if gt.shape[-2:] != size:
        gt = F.adaptive_avg_pool2d(gt.float(), size).to(pred.dtype)
    m = mask.float()
    if m.shape[-2:] != size:
        m = F.interpolate(m, size=size, mode="bilinear", align_corners=False)
    mask = ((m >= 0.5).float() * m).to(pred.dtype)
    cw = torch.ones(C, dtype=pred.dtype, device=pred.device)
    cw[heat_start:bkg_start] = keypoint_task_weight
    cw[C - 2] = multi_task_weight
    gt_b = gt.unsqueeze(0)
    mask_b = (mask.unsqueeze(0) * cw.view(1, 1, C, 1, 1))
    st = torch.where(gt_b >= 0.01, pred - alpha, 1.0 - pred - beta)
    if gamma == 1:
        factor = torch.abs(1.0 - st)
    else:
        factor = (1.0 - st) ** gamma
    out = (pred - gt_b) ** 2 * factor * mask_b
    loss_nstack = out.sum(dim=(1, 2, 3, 4))
    w = [loss_nstack[i] * nstack_weight[i] for i in range(nstack)]
    return sum(w) / sum(nstack_weight)


# ---------------------------------------------------------------------------
# post-processing primitives (device path wired up in ops/postproc.py)
# ---------------------------------------------------------------------------

def heatmap_nms(heat, threshold=0.1):
    """3x3 max-pool equality peak mask (reference utils/util.py:177-183)."""
    if use_hip_for(heat):
        from . import postproc
        return postproc.heatmap_nms_hip(heat, threshold)
    maxm = F.max_pool2d(heat, 3, 1, 1)
    maxm = torch.eq(maxm, heat).to(heat.dtype)
    return heat * maxm * (heat > threshold).to(heat.dtype)
