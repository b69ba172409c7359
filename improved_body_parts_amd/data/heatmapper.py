"""Ground-truth heatmap generation (CPU/numpy oracle).

Capability parity with reference py_cocodata_server/py_data_heatmapper.py
(Heatmapper :8-299, distances :309-340): Gaussian keypoint maps sampled on the
original-resolution grid every ``stride`` px (quantisation-free trick,
reference :104-155), limb ("PAF-as-Gaussian") maps from perpendicular
distance-to-segment with per-pixel hit-count averaging (reference :163-240),
person-mask background (eroded mask_all) and reverse-keypoint background
channels (reference :74-80), optional shared offset maps (reference :242-299).

This module is the NUMERICS ORACLE for the on-device HIP GT generator
(ops/csrc/heatmap_gt.hip); the device path is what training uses on MI355X.
Implementation is vectorised numpy, not a loop port.
"""
from __future__ import annotations

from math import ceil, log, sqrt

import numpy as np


def _erode3x3(mask: np.ndarray) -> np.ndarray:
    """Binary-style 3x3 erosion (min filter) with edge replication, matching
    cv2.erode(mask, ones((3,3))) semantics on float input."""
    p = np.pad(mask, 1, mode="edge")
    out = mask.copy()
    for dy in (0, 1, 2):
        for dx in (0, 1, 2):
            np.minimum(out, p[dy:dy + mask.shape[0], dx:dx + mask.shape[1]], out=out)
    return out


class Heatmapper:
    def __init__(self, config):
        self.config = config
        tp = config.transform_params
        self.sigma = tp.sigma
        self.paf_sigma = tp.paf_sigma
        self.double_sigma2 = 2.0 * self.sigma * self.sigma
        self.keypoint_gaussian_thre = tp.keypoint_gaussian_thre
        self.limb_gaussian_thre = tp.limb_gaussian_thre
        self.gaussian_size = ceil(sqrt(-self.double_sigma2 * log(self.keypoint_gaussian_thre))
                                  / config.stride) * 2
        self.offset_size = self.gaussian_size // 2 + 1
        self.thre = tp.paf_thre

        stride = config.stride
        w = config.width // stride
        h = config.height // stride
        # original-resolution coordinates of each output-grid cell centre
        self.grid_x = (np.arange(w) * stride + stride / 2 - 0.5).astype(np.float32)
        self.grid_y = (np.arange(h) * stride + stride / 2 - 0.5).astype(np.float32)
        self.X, self.Y = np.meshgrid(self.grid_x, self.grid_y)

    # ------------------------------------------------------------------ public
    def create_heatmaps(self, joints: np.ndarray, mask_all: np.ndarray) -> np.ndarray:
        """joints: (P, num_parts, 3) canonical-order; mask_all: (H/stride, W/stride).
        Returns (num_layers, H/stride, W/stride) float32 in [0, 1] (CHW)."""
        cfg = self.config
        maps = np.zeros(cfg.parts_shape, dtype=np.float32)  # HWC
        self.put_joints(maps, joints)
        self.put_limbs(maps, joints)
        # background channel 1: eroded all-person mask (reference :74-76)
        maps[:, :, cfg.bkg_start] = _erode3x3(mask_all.astype(np.float32))
        # background channel 2: max over all keypoint channels (reference :79-80)
        sl = slice(cfg.heat_start, cfg.heat_start + cfg.heat_layers)
        maps[:, :, cfg.bkg_start + 1] = np.amax(maps[:, :, sl], axis=2)
        np.clip(maps, 0.0, 1.0, out=maps)
        return maps.transpose((2, 0, 1))

    # ---------------------------------------------------------------- keypoints
    def put_joints(self, heatmaps, joints):
        for part in range(self.config.num_parts):
            visible = joints[:, part, 2] < 2
            self.put_gaussian_maps(heatmaps, part, joints[visible, part, 0:2])

    def put_gaussian_maps(self, heatmaps, layer, pts):
        """Separable-Gaussian keypoint response, max-combined over persons
        (reference :99-155). The window is ±gaussian_size/2 grid cells around the
        rounded keypoint location."""
        cfg = self.config
        half = self.gaussian_size // 2
        ch = cfg.heat_start + layer
        H, W = heatmaps.shape[:2]
        for x, y in np.asarray(pts, dtype=np.float32):
            cx = int(round(x / cfg.stride))
            cy = int(round(y / cfg.stride))
            x0, x1 = max(cx - half, 0), min(cx + half + 1, W)
            y0, y1 = max(cy - half, 0), min(cy + half + 1, H)
            if x1 <= x0 or y1 <= y0:
                continue
            ex = np.exp(-((self.grid_x[x0:x1] - x) ** 2) / np.float32(self.double_sigma2))
            ey = np.exp(-((self.grid_y[y0:y1] - y) ** 2) / np.float32(self.double_sigma2))
            np.maximum(heatmaps[y0:y1, x0:x1, ch], np.outer(ey, ex),
                       out=heatmaps[y0:y1, x0:x1, ch])

    # -------------------------------------------------------------------- limbs
    def put_limbs(self, heatmaps, joints):
        for i, (fr, to) in enumerate(self.config.limbs_conn):
            visible = (joints[:, fr, 2] < 2) & (joints[:, to, 2] < 2)
            layer = self.config.paf_start + i
            self.put_limb_gaussian_maps(heatmaps, layer,
                                        joints[visible, fr, 0:2], joints[visible, to, 0:2])

    def put_limb_gaussian_maps(self, heatmaps, layer, joint_from, joint_to):
        """Gaussian-of-perpendicular-distance limb response, averaged by hit count
        (reference :163-227). Responses below the threshold contribute 0.01."""
        cfg = self.config
        H, W = heatmaps.shape[:2]
        acc = heatmaps[:, :, layer]
        count = np.zeros((H, W), dtype=np.float32)
        for (x1, y1), (x2, y2) in zip(np.asarray(joint_from, np.float32),
                                      np.asarray(joint_to, np.float32)):
            dx, dy = x2 - x1, y2 - y1
            dnorm2 = dx * dx + dy * dy
            if dnorm2 == 0:
                continue  # coincident end-points produce no limb (reference :177-182)
            # bounding box in grid cells, dilated by paf_thre original px
            min_sx = int(round((min(x1, x2) - self.thre) / cfg.stride))
            min_sy = int(round((min(y1, y2) - self.thre) / cfg.stride))
            max_sx = int(round((max(x1, x2) + self.thre) / cfg.stride))
            max_sy = int(round((max(y1, y2) + self.thre) / cfg.stride))
            if max_sx < 0 or max_sy < 0:
                continue
            x0g, y0g = max(min_sx, 0), max(min_sy, 0)
            x1g, y1g = min(max_sx + 1, W), min(max_sy + 1, H)
            if x1g <= x0g or y1g <= y0g:
                continue
            d = limb_gaussian(self.X[y0g:y1g, x0g:x1g], self.Y[y0g:y1g, x0g:x1g],
                              self.paf_sigma, x1, y1, x2, y2, self.limb_gaussian_thre)
            hit = d > 0
            acc[y0g:y1g, x0g:x1g][hit] += d[hit]
            count[y0g:y1g, x0g:x1g][hit] += 1
        nz = count > 0
        acc[nz] /= count[nz]

    # ------------------------------------------------------------------ offsets
    def put_offset(self, joints):
        """Shared x/y offset maps + mask (reference :242-299). Disabled in the
        default training path, kept for capability parity."""
        cfg = self.config
        off = np.zeros(cfg.offset_shape, dtype=np.float32)
        cnt = np.zeros(cfg.offset_shape, dtype=np.float32)
        half = self.offset_size // 2
        H, W = cfg.offset_shape[:2]
        denom = self.offset_size * cfg.stride
        for part in range(cfg.num_parts):
            visible = joints[:, part, 2] < 2
            for x, y in joints[visible, part, 0:2]:
                cx, cy = int(round(x / cfg.stride)), int(round(y / cfg.stride))
                x0, x1 = max(cx - half, 0), min(cx + half + 1, W)
                y0, y1 = max(cy - half, 0), min(cy + half + 1, H)
                if x1 <= x0 or y1 <= y0:
                    continue
                ox = (self.grid_x[x0:x1] - np.float32(x)) / denom
                oy = (self.grid_y[y0:y1] - np.float32(y)) / denom
                off[y0:y1, x0:x1, 0] += ox[None, :]
                off[y0:y1, x0:x1, 1] += oy[:, None]
                cnt[y0:y1, x0:x1, :] += 1
        nz = cnt > 0
        off[nz] /= cnt[nz]
        mask = np.zeros_like(cnt)
        mask[nz] = 1
        return off.transpose((2, 0, 1)), mask.transpose((2, 0, 1))


def create_heatmaps_device(joints_batch, mask_all_batch, config, device="cuda"):
    """On-device batched GT generation (HIP kernel csrc/heatmap_gt.hip).

    joints_batch: (N, P, num_parts, 3) array/tensor, original-resolution
    coordinates, visibility 2 = absent (pad rows with vis=2).
    mask_all_batch: (N, h, w) stride-grid masks or None.
    Returns (N, num_layers, h, w) fp32 CUDA tensor — numerically matching the
    CPU oracle ``Heatmapper.create_heatmaps`` (tested in tests/test_ops_gpu).
    """
    import torch as _torch

    from ..ops._backend import require_hip

    ext = require_hip()
    assert config.paf_start == 0, "device GT kernel assumes PAF-first layout"
    tp = config.transform_params
    h = config.height // config.stride
    w = config.width // config.stride
    joints = _torch.as_tensor(np.asarray(joints_batch, dtype=np.float32)
                              if not isinstance(joints_batch, _torch.Tensor)
                              else joints_batch, dtype=_torch.float32,
                              device=device)
    mask = None
    if mask_all_batch is not None:
        mask = _torch.as_tensor(np.asarray(mask_all_batch, dtype=np.float32)
                                if not isinstance(mask_all_batch, _torch.Tensor)
                                else mask_all_batch, dtype=_torch.float32,
                                device=device)
    limb_pairs = _torch.tensor(np.asarray(config.limbs_conn, dtype=np.int32),
                               dtype=_torch.int32)
    return ext.heatmap_gt(joints, mask, limb_pairs, h, w, config.stride,
                          config.heat_start, config.bkg_start,
                          config.num_layers, tp.sigma, tp.paf_sigma,
                          tp.keypoint_gaussian_thre, tp.limb_gaussian_thre,
                          tp.paf_thre)


def limb_gaussian(X, Y, sigma, x1, y1, x2, y2, thresh=0.01):
    """Gaussian of the perpendicular distance from grid points to the segment's
    carrier line (reference distances() :309-340, including its quirk of writing
    0.01 where the response is at or below the threshold)."""
    xD, yD = x2 - x1, y2 - y1
    norm = sqrt(xD * xD + yD * yD)
    dist = np.abs((xD * (y1 - Y) - (x1 - X) * yD) / (norm + 1e-6))
    g = np.exp(-(dist * dist) / (2.0 * sigma * sigma))
    g[g <= thresh] = 0.01
    return g.astype(np.float32)
