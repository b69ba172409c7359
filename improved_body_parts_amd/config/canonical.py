"""Canonical skeleton / channel-layout configuration.

Capability parity with reference config/config.py (CanonicalConfig at :52-134,
TrainingOpt :8-23, TransformationParams :26-49, COCOSourceConfig :137-233,
GetConfig :243-260), re-designed:

  * one parameterised ``CanonicalConfig`` instead of four forked config files
    (reference config2.py / config_dense.py / config_final.py are whole-file copies);
  * no print side effects in ``GetConfig``;
  * channel layout identical to the reference so checkpoints and post-processing
    interoperate: PAF channels first (0..29), keypoint channels next (30..47),
    then person-mask background and reverse-keypoint background (48, 49).
"""
from __future__ import annotations

import numpy as np


class TrainingOpt:
    """Training hyper-parameters (defaults match reference config/config.py:8-23)."""

    def __init__(self, **overrides):
        self.batch_size = 4            # per process; global batch = batch_size * world_size
        self.learning_rate = 2.5e-5    # per process; scaled by world_size in distributed training
        self.config_name = "Canonical"
        self.hdf5_train_data = "./data/dataset/coco/link2coco2017/coco_train_dataset512.h5"
        self.hdf5_val_data = "./data/dataset/coco/link2coco2017/coco_val_dataset512.h5"
        self.nstack = 4                # number of stacked hourglasses
        self.model_variant = "imhn"    # imhn | final | attention | light | independent | ae
        self.hourglass_inp_dim = 256   # channels entering each hourglass
        self.increase = 128            # channel increase per down-sampling inside the hourglass
        self.nstack_weight = [1, 1, 1, 1]
        self.scale_weight = [0.1, 0.2, 0.4, 1.6, 6.4]  # scales 128,64,32,16,8
        self.multi_task_weight = 0.1   # person-mask channel loss weight
        self.keypoint_task_weight = 3  # keypoint channels loss weight (vs body-part channels)
        self.ckpt_path = "./checkpoints/PoseNet_52_epoch.pth"
        # Native additions (no Apex in this framework):
        self.dtype = "bf16"            # compute dtype on MI355X; master weights stay fp32
        self.momentum = 0.9
        self.weight_decay = 2e-4
        self.warmup_epochs = 3
        self.lr_decay_every = 15       # epochs between x0.2 decays (reference train_distributed.py:382-400)
        self.lr_decay_factor = 0.2
        # drop batches above this (reference train_distributed.py:259-261 used
        # 2e5 for sparse COCO crops; dense synthetic scenes start near 2e5 at
        # random init, so the guard sits well above the healthy range)
        self.loss_explosion_thre = 1e7
        for k, v in overrides.items():
            if not hasattr(self, k):
                raise AttributeError(f"unknown TrainingOpt field {k!r}")
            setattr(self, k, v)
        if len(self.nstack_weight) != self.nstack:
            self.nstack_weight = [1] * self.nstack


class TransformationParams:
    """Augmentation / GT-generation hyper-parameters (reference config/config.py:26-49)."""

    def __init__(self, stride: int):
        self.target_dist = 0.6
        self.scale_prob = 0.8
        self.scale_min = 0.7
        self.scale_max = 1.3
        self.max_rotate_degree = 40.0
        self.center_perterb_max = 50.0
        self.flip_prob = 0.5
        self.tint_prob = 0.2
        self.sigma = 9                  # keypoint Gaussian sigma (512 input)
        self.keypoint_gaussian_thre = 0.015
        self.limb_gaussian_thre = 0.015
        self.paf_sigma = 7              # body-part (limb) Gaussian sigma
        self.paf_thre = 1.0 * stride    # limb bounding-box dilation to include end-points


# The 18 canonical parts (reference config/config.py:61-62). Order is load-bearing:
# channel layout, flip tables and the COCO adapter all index into it.
_PARTS = [
    "nose", "neck", "Rsho", "Relb", "Rwri", "Lsho", "Lelb", "Lwri",
    "Rhip", "Rkne", "Rank", "Lhip", "Lkne", "Lank", "Reye", "Leye", "Rear", "Lear",
]

# 30 limb connections by part name (reference config/config.py:74-82).
_LIMBS = [
    ("neck", "nose"), ("neck", "Reye"), ("neck", "Leye"), ("neck", "Rear"), ("neck", "Lear"),
    ("nose", "Reye"), ("nose", "Leye"), ("Reye", "Rear"), ("Leye", "Lear"),
    ("neck", "Rsho"), ("Rsho", "Relb"), ("Relb", "Rwri"),
    ("neck", "Lsho"), ("Lsho", "Lelb"), ("Lelb", "Lwri"),
    ("neck", "Rhip"), ("Rhip", "Rkne"), ("Rkne", "Rank"),
    ("neck", "Lhip"), ("Lhip", "Lkne"), ("Lkne", "Lank"),
    ("nose", "Rsho"), ("nose", "Lsho"), ("Rsho", "Rhip"), ("Rhip", "Lkne"),
    ("Lsho", "Lhip"), ("Lhip", "Rkne"), ("Rear", "Rsho"), ("Lear", "Lsho"), ("Rhip", "Lhip"),
]

# Dense-skeleton variant (reference config_dense.py): every-pair style extension
# adding 19 more connections for a total of 49.
_DENSE_EXTRA_LIMBS = [
    ("nose", "Rwri"), ("nose", "Lwri"), ("nose", "Rhip"), ("nose", "Lhip"),
    ("Rsho", "Lhip"), ("Lsho", "Rhip"), ("Rsho", "Rkne"), ("Lsho", "Lkne"),
    ("Relb", "Rhip"), ("Lelb", "Lhip"), ("Rwri", "Rhip"), ("Lwri", "Lhip"),
    ("Rhip", "Rank"), ("Lhip", "Lank"), ("neck", "Rkne"), ("neck", "Lkne"),
    ("Rear", "Lear"), ("Reye", "Leye"), ("Rsho", "Lsho"),
]


class CanonicalConfig:
    """Skeleton definition + channel layout (reference config/config.py:52-134).

    Channel layout (num_layers = paf_layers + heat_layers + 2):
      [0, paf_layers)                   limb ("PAF-as-Gaussian") channels
      [heat_start, heat_start+18)       keypoint channels
      [bkg_start]                       person-mask background
      [bkg_start+1]                     reverse-keypoint background
    """

    def __init__(self, width: int = 512, height: int = 512, stride: int = 4,
                 limbs=None):
        self.width = width
        self.height = height
        self.stride = stride

        self.parts = list(_PARTS)
        self.num_parts = len(self.parts)
        self.parts_dict = {p: i for i, p in enumerate(self.parts)}
        # background pseudo-parts appended after the dict is frozen (reference :67-70)
        self.parts = self.parts + ["background", "reverseKeypoint"]
        self.num_parts_with_background = len(self.parts)

        self.leftParts = [self.parts_dict[p] for p in
                          ["Lsho", "Lelb", "Lwri", "Lhip", "Lkne", "Lank", "Leye", "Lear"]]
        self.rightParts = [self.parts_dict[p] for p in
                           ["Rsho", "Relb", "Rwri", "Rhip", "Rkne", "Rank", "Reye", "Rear"]]

        limb_names = list(_LIMBS) if limbs is None else list(limbs)
        self.limb_from = [self.parts_dict[a] for a, _ in limb_names]
        self.limb_to = [self.parts_dict[b] for _, b in limb_names]
        self.limbs_conn = list(zip(self.limb_from, self.limb_to))

        self.paf_layers = len(self.limbs_conn)
        self.heat_layers = self.num_parts
        self.num_layers = self.paf_layers + self.heat_layers + 2

        self.paf_start = 0
        self.heat_start = self.paf_layers
        self.bkg_start = self.paf_layers + self.heat_layers

        self.offset_layers = 2
        self.offset_start = self.num_layers

        self.mask_shape = (height // stride, width // stride)
        self.parts_shape = (height // stride, width // stride, self.num_layers)
        self.offset_shape = (height // stride, width // stride, self.offset_layers)

        self.transform_params = TransformationParams(stride)

        # COCO detection-id -> ground-truth-id map used when exporting results
        # (reference config/config.py:117-118).
        self.dt_gt_mapping = {0: 0, 1: None, 2: 6, 3: 8, 4: 10, 5: 5, 6: 7, 7: 9,
                              8: 12, 9: 14, 10: 16, 11: 11, 12: 13, 13: 15,
                              14: 2, 15: 1, 16: 4, 17: 3}

        self.flip_heat_ord = self._build_flip_heat_order()
        self.flip_paf_ord = self._build_flip_paf_order()
        self.draw_list = [0] + list(range(5, 21)) + [29] if self.paf_layers >= 30 else list(range(self.paf_layers))

    # -- flip permutation tables -------------------------------------------------
    # The reference hard-codes these (config/config.py:121-124); we derive them from
    # the part/limb tables so config variants stay consistent automatically.
    def _flip_part(self, idx: int) -> int:
        if idx in self.leftParts:
            return self.rightParts[self.leftParts.index(idx)]
        if idx in self.rightParts:
            return self.leftParts[self.rightParts.index(idx)]
        return idx

    def _build_flip_heat_order(self) -> np.ndarray:
        # keypoint channels + the two background channels keep their slots
        order = [self._flip_part(i) for i in range(self.num_parts)]
        order += [self.num_parts, self.num_parts + 1]  # background channels unswapped
        return np.array(order, dtype=np.int64)

    def _build_flip_paf_order(self) -> np.ndarray:
        flipped_pairs = [tuple(sorted((self._flip_part(a), self._flip_part(b))))
                         for (a, b) in self.limbs_conn]
        orig_pairs = [tuple(sorted(p)) for p in self.limbs_conn]
        order = []
        for fp in flipped_pairs:
            order.append(orig_pairs.index(fp))
        return np.array(order, dtype=np.int64)


def DenseSkeletonConfig(width: int = 512, height: int = 512, stride: int = 4) -> CanonicalConfig:
    """49-limb dense-skeleton variant (capability of reference config_dense.py)."""
    return CanonicalConfig(width, height, stride, limbs=_LIMBS + _DENSE_EXTRA_LIMBS)


# 24-limb slim skeleton of the 3-stage @384 variant (reference config2.py:69-83)
_SLIM_LIMBS = [
    ("neck", "nose"), ("neck", "Reye"), ("neck", "Leye"), ("neck", "Rear"),
    ("neck", "Lear"), ("nose", "Reye"), ("nose", "Leye"), ("Reye", "Rear"),
    ("Leye", "Lear"), ("neck", "Rsho"), ("Rsho", "Relb"), ("Relb", "Rwri"),
    ("neck", "Lsho"), ("Lsho", "Lelb"), ("Lelb", "Lwri"), ("neck", "Rhip"),
    ("Rhip", "Rkne"), ("Rkne", "Rank"), ("neck", "Lhip"), ("Lhip", "Lkne"),
    ("Lkne", "Lank"), ("Rhip", "Lhip"), ("Rsho", "Rear"), ("Lsho", "Lear"),
]


def SlimSkeletonConfig(width: int = 384, height: int = 384, stride: int = 4) -> CanonicalConfig:
    """24-limb 44-channel variant at 384^2 (capability of reference
    config2.py's 3-stage configuration)."""
    return CanonicalConfig(width, height, stride, limbs=_SLIM_LIMBS)


class COCOSourceConfig:
    """COCO dataset joint order -> canonical order adapter (reference config/config.py:137-233)."""

    def __init__(self, hdf5_source: str = ""):
        self.hdf5_source = hdf5_source
        self.parts = ["nose", "Leye", "Reye", "Lear", "Rear", "Lsho", "Rsho", "Lelb",
                      "Relb", "Lwri", "Rwri", "Lhip", "Rhip", "Lkne", "Rkne", "Lank", "Rank"]
        self.num_parts = len(self.parts)
        self.parts_dict = {p: i for i, p in enumerate(self.parts)}

    def convert(self, meta: dict, global_config: CanonicalConfig) -> dict:
        """Convert COCO-order joints to canonical order, synthesising the neck.

        Visibility convention (matches reference): 0 = marked but invisible,
        1 = marked and visible, 2 = not marked on this person,
        3 = never marked in this dataset.
        """
        joints = np.array(meta["joints"], dtype=np.float32)
        assert joints.shape[1] == self.num_parts
        out = np.zeros((joints.shape[0], global_config.num_parts, 3), dtype=np.float32)
        out[:, :, 2] = 3.0
        for p in self.parts:
            if p in global_config.parts_dict:
                gid = global_config.parts_dict[p]
                out[:, gid, :] = joints[:, self.parts_dict[p], :]
        # synthesise the neck as the mean of the shoulders when both are marked
        neck = global_config.parts_dict["neck"]
        r, l = self.parts_dict["Rsho"], self.parts_dict["Lsho"]
        known = (joints[:, l, 2] < 2) & (joints[:, r, 2] < 2)
        out[~known, neck, 2] = 2.0
        out[known, neck, 0:2] = (joints[known, r, 0:2] + joints[known, l, 0:2]) / 2
        out[known, neck, 2] = np.minimum(joints[known, r, 2], joints[known, l, 2])
        meta = dict(meta)
        meta["joints"] = out
        return meta

    def repeat_mask(self, mask, global_config, joints=None):
        return np.repeat(mask[:, :, np.newaxis], global_config.num_layers, axis=2)

    def source(self) -> str:
        return self.hdf5_source


# Registry (reference config/config.py:239-260). Entries are zero-arg factories.
Configs = {
    "Canonical": CanonicalConfig,
    "Canonical384": lambda: CanonicalConfig(384, 384, 4),
    "Canonical768": lambda: CanonicalConfig(768, 768, 4),
    "DenseSkeleton": DenseSkeletonConfig,
    "Slim384": SlimSkeletonConfig,
}


def GetConfig(config_name: str) -> CanonicalConfig:
    config = Configs[config_name]()
    # invariants the reference asserts at import time (config/config.py:87-92)
    assert config.num_layers == config.paf_layers + config.heat_layers + 2
    assert len(config.flip_paf_ord) == config.paf_layers
    assert len(config.flip_heat_ord) == config.heat_layers + 2
    return config
