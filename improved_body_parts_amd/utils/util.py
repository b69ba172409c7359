"""Utility helpers (capability parity with reference utils/util.py).

padRightDownCorner/center_pad (reference :44-100), GaussianSmoothing (:103-174),
keypoint_heatmap_nms (:177-183), refine_centroid (:186-211; the reference's
version mixes up the x/y meshgrids — ours uses the correct axes, which the
reference's own comment says makes no measurable difference), set_bn_eval
(:214-223, bf16 here instead of fp16), plus the training meters/LR schedule the
reference keeps in its drivers.
"""
from __future__ import annotations

import math
import numbers

import numpy as np
import torch
import torch.nn.functional as F
from torch import nn


def padRightDownCorner(img, stride, padValue):
    """Pad (H, W, C) image on the bottom/right to a multiple of ``stride``."""
    h, w = img.shape[0], img.shape[1]
    pad = [0, 0,
           0 if h % stride == 0 else stride - (h % stride),
           0 if w % stride == 0 else stride - (w % stride)]
    out = np.pad(img, ((pad[0], pad[2]), (pad[1], pad[3]), (0, 0)),
                 mode="constant", constant_values=padValue)
    return out, pad


def center_pad(image, stride, padValue):
    """Pad (H, W, C) image symmetrically to a multiple of ``stride``."""
    h, w = image.shape[0], image.shape[1]
    dh = 0 if h % stride == 0 else stride - (h % stride)
    dw = 0 if w % stride == 0 else stride - (w % stride)
    top, left = dh // 2, dw // 2
    pad = [top, left, dh - top, dw - left]  # up, left, down, right
    out = np.pad(image, ((pad[0], pad[2]), (pad[1], pad[3]), (0, 0)),
                 mode="constant", constant_values=padValue)
    return out, pad


class GaussianSmoothing(nn.Module):
    """Depthwise Gaussian smoothing, 1/2/3-D (reference utils/util.py:103-174)."""

    def __init__(self, channels, kernel_size, sigma, dim=2):
        super().__init__()
        if isinstance(kernel_size, numbers.Number):
            kernel_size = [kernel_size] * dim
        if isinstance(sigma, numbers.Number):
            sigma = [sigma] * dim
        kernel = torch.ones(1)
        meshgrids = torch.meshgrid(
            [torch.arange(s, dtype=torch.float32) for s in kernel_size], indexing="ij")
        for size, std, mgrid in zip(kernel_size, sigma, meshgrids):
            mean = (size - 1) / 2
            kernel = kernel * (1 / (std * math.sqrt(2 * math.pi))
                               * torch.exp(-(((mgrid - mean) / std) ** 2) / 2))
        kernel = kernel / kernel.sum()
        # NOTE: view first, THEN take dim() — evaluating kernel.dim() inside
        # the same statement reads the pre-view tensor (latent crash for any
        # dim, caught by tests/test_utils.py)
        kernel = kernel.view(1, 1, *kernel.shape)
        kernel = kernel.repeat(channels, *([1] * (kernel.dim() - 1)))
        self.register_buffer("weight", kernel)
        self.groups = channels
        self.conv = {1: F.conv1d, 2: F.conv2d, 3: F.conv3d}[dim]

    def forward(self, x):
        return self.conv(x, weight=self.weight.to(x.device, x.dtype), groups=self.groups)


def keypoint_heatmap_nms(heat, kernel=3, thre=0.1):
    """Peak mask via max-pool equality (reference utils/util.py:177-183)."""
    pad = (kernel - 1) // 2
    pad_heat = F.pad(heat, (pad, pad, pad, pad), mode="reflect")
    hmax = F.max_pool2d(pad_heat, (kernel, kernel), stride=1, padding=0)
    keep = (hmax == heat).to(heat.dtype) * (heat >= thre).to(heat.dtype)
    return heat * keep


def refine_centroid(scorefmp, anchor, radius):
    """Sub-pixel refinement by weighted centroid over a (2r+1)^2 box
    (reference utils/util.py:186-211)."""
    x_c, y_c = anchor
    x_min, x_max = x_c - radius, x_c + radius + 1
    y_min, y_max = y_c - radius, y_c + radius + 1
    if y_max > scorefmp.shape[0] or y_min < 0 or x_max > scorefmp.shape[1] or x_min < 0:
        return tuple(anchor) + (scorefmp[y_c, x_c],)
    box = scorefmp[y_min:y_max, x_min:x_max]
    y_grid, x_grid = np.mgrid[-radius:radius + 1, -radius:radius + 1]
    s = box.sum()
    return (x_c + (box * x_grid).sum() / s,
            y_c + (box * y_grid).sum() / s,
            box.mean())


def set_bn_eval_fp32(m):
    if "BatchNorm" in m.__class__.__name__:
        m.eval()


def set_bn_eval(m):
    """Freeze BN and run it in bf16 (SWA fine-tune path; the reference used fp16)."""
    if "BatchNorm" in m.__class__.__name__:
        m.eval().bfloat16()


class AverageMeter:
    """Running average tracker (reference train_distributed.py AverageMeter)."""

    def __init__(self):
        self.reset()

    def reset(self):
        self.val = 0.0
        self.avg = 0.0
        self.sum = 0.0
        self.count = 0

    def update(self, val, n=1):
        self.val = val
        self.sum += val * n
        self.count += n
        self.avg = self.sum / max(self.count, 1)


def adjust_learning_rate(optimizer, epoch, iters_done, iters_per_epoch, base_lr,
                         warmup_epochs=3, decay_every=15, decay_factor=0.2):
    """Per-iteration LR schedule: linear warm-up over ``warmup_epochs`` epochs,
    then x``decay_factor`` every ``decay_every`` epochs
    (reference train_distributed.py:382-400)."""
    lr = base_lr * (decay_factor ** (epoch // decay_every))
    if epoch < warmup_epochs and iters_per_epoch > 0:
        progress = (epoch * iters_per_epoch + iters_done) / (warmup_epochs * iters_per_epoch)
        lr = lr * min(1.0, max(progress, 1e-4))
    for group in optimizer.param_groups:
        group["lr"] = lr
    return lr
