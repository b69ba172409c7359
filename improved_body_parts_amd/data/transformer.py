"""Augmentation: one combined affine for image, masks and joints.

Capability parity with reference py_cocodata_server/py_data_transformer.py
(AugmentSelection :14-89, Transformer :92-184): a single affine matrix composed
of center -> rotate -> scale -> flip -> re-center(+jitter) is applied once to the
image and both masks; joints go through the same matrix as homogeneous points;
left/right part ids swap on horizontal flip; masks are warped then resized to
the stride-4 output grid; output image is float32 in [0, 1].

Re-designed: cv2 is not a dependency — image warping uses
``scipy.ndimage.affine_transform`` (order-1, matching cv2.warpAffine INTER_LINEAR
semantics) and the mask downscale is an area-mean pool (INTER_AREA equivalent).
"""
from __future__ import annotations

import random
from math import cos, sin, pi

import numpy as np
from scipy import ndimage


class AugmentSelection:
    def __init__(self, flip=False, degree=0.0, crop=(0, 0), scale=1.0, tint=False):
        self.flip = flip
        self.degree = degree
        self.crop = crop
        self.scale = scale
        self.tint = tint

    @classmethod
    def random(cls, transform_params, rng: random.Random | None = None):
        rng = rng or random
        tp = transform_params
        flip = rng.uniform(0, 1) < tp.flip_prob
        degree = rng.uniform(-1, 1) * tp.max_rotate_degree
        scale = (rng.uniform(tp.scale_min, tp.scale_max)
                 if rng.uniform(0, 1) < tp.scale_prob else 1.0)
        x_off = int(rng.uniform(-1, 1) * tp.center_perterb_max)
        y_off = int(rng.uniform(-1, 1) * tp.center_perterb_max)
        tint = rng.uniform(0, 1) < tp.tint_prob
        return cls(flip, degree, (x_off, y_off), scale, tint)

    @classmethod
    def unrandom(cls):
        return cls(False, 0.0, (0, 0), 1.0, False)

    def affine(self, center, scale_self, config):
        """Compose the 2x3 affine mapping source -> destination (reference :43-89).

        ``scale_self`` is the main person's height fraction; the image is scaled
        so that person height becomes ``target_dist`` of the crop.
        """
        tp = config.transform_params
        scale_size = tp.target_dist / max(scale_self, 1e-6) * self.scale
        deg = self.degree * pi / 180.0
        A = scale_size * cos(deg)
        B = scale_size * sin(deg)
        w, h = config.width, config.height
        (cx, cy) = center
        cx += self.crop[0]
        cy += self.crop[1]
        # rotate+scale about the (jittered) person center, then translate that
        # center to the crop center; flip mirrors x about the crop center.
        flip_sign = -1.0 if self.flip else 1.0
        m = np.array([
            [flip_sign * A, flip_sign * B, w / 2 - flip_sign * (A * cx + B * cy)],
            [-B, A, h / 2 - (-B * cx + A * cy)],
        ], dtype=np.float32)
        return m


class Transformer:
    def __init__(self, config):
        self.config = config

    def transform(self, img, mask_miss, mask_all, meta, aug=None, rng=None):
        """Apply one combined affine to image + masks + joints.

        :param img: (H, W, 3) uint8 or float
        :param mask_miss, mask_all: (H, W) float/uint8 in [0,1] (or [0,255])
        :param meta: dict with 'objpos' (main-person center) and 'joints'
            (P, num_parts, 3) canonical-order
        :returns: (img float32 [0,1] (H,W,3), mask_miss (h,w), mask_all (h,w), meta)
        """
        cfg = self.config
        aug = aug or AugmentSelection.random(cfg.transform_params, rng)
        scale_self = meta.get("scale_provided", 1.0)
        center = np.asarray(meta["objpos"], dtype=np.float32).reshape(2)
        M = aug.affine(center, scale_self, cfg)

        img = self._warp(img, M, (cfg.height, cfg.width), order=1,
                         cval=float(np.mean((128,))))
        mask_miss = self._warp(self._to_float01(mask_miss), M, (cfg.height, cfg.width),
                               order=1, cval=1.0)
        mask_all = self._warp(self._to_float01(mask_all), M, (cfg.height, cfg.width),
                              order=1, cval=0.0)

        joints = np.array(meta["joints"], dtype=np.float32, copy=True)
        pts = np.concatenate([joints[:, :, 0:2],
                              np.ones(joints.shape[:2] + (1,), np.float32)], axis=2)
        joints[:, :, 0:2] = pts @ M.T
        if aug.flip:
            # swap left/right part ids (reference :173-177)
            order = np.arange(cfg.num_parts)
            for l, r in zip(cfg.leftParts, cfg.rightParts):
                order[l], order[r] = r, l
            joints = joints[:, order, :]
        meta = dict(meta)
        meta["joints"] = joints

        if aug.tint and img.ndim == 3:
            img = self._tint(img, rng)

        # masks live on the stride-grid (reference :178-183)
        s = cfg.stride
        mask_miss = self._area_pool(mask_miss, s)
        mask_all = self._area_pool(mask_all, s)

        img = np.ascontiguousarray(img, dtype=np.float32)
        if img.max() > 1.5:
            img = img / 255.0
        return img, mask_miss.astype(np.float32), mask_all.astype(np.float32), meta

    # ------------------------------------------------------------------ helpers
    @staticmethod
    def _to_float01(m):
        m = np.asarray(m, dtype=np.float32)
        if m.max() > 1.5:
            m = m / 255.0
        return m

    @staticmethod
    def _warp(img, M, out_shape, order=1, cval=0.0):
        """Apply the dst = M @ src affine. scipy wants the inverse (dst -> src)."""
        Mi = np.linalg.inv(np.vstack([M, [0, 0, 1]]))[:2]
        # scipy affine_transform maps output coords o to input via matrix @ o + offset,
        # in (row, col) = (y, x) order.
        mat = Mi[[1, 0]][:, [1, 0]]  # swap x/y axes to (y, x) convention
        offset = Mi[[1, 0], 2]
        if img.ndim == 2:
            return ndimage.affine_transform(img, mat, offset=offset,
                                            output_shape=out_shape, order=order,
                                            mode="constant", cval=cval)
        chans = [ndimage.affine_transform(img[..., c].astype(np.float32), mat,
                                          offset=offset, output_shape=out_shape,
                                          order=order, mode="constant", cval=cval)
                 for c in range(img.shape[-1])]
        return np.stack(chans, axis=-1)

    @staticmethod
    def _area_pool(m, s):
        h, w = m.shape
        return m[:h - h % s, :w - w % s].reshape(h // s, s, w // s, s).mean(axis=(1, 3))

    @staticmethod
    def _tint(img, rng=None):
        """HSV-ish tint distortion (reference :98-110) without cv2: random
        per-channel gain + brightness shift in RGB."""
        rng = rng or random
        img = img.astype(np.float32)
        if img.max() > 1.5:
            img = img / 255.0
        gains = np.array([1.0 + rng.uniform(-0.15, 0.15) for _ in range(3)], np.float32)
        shift = rng.uniform(-0.1, 0.1)
        return np.clip(img * gains + shift, 0.0, 1.0)
