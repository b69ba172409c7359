"""COCO HDF5 dataset path (capability parity with reference data/mydataset.py,
py_cocodata_server/py_data_iterator.py and data/coco_masks_hdf5.py).

The build environment has no h5py / pycocotools / network, so this module
import-gates those dependencies: constructing the classes without them raises a
clear error, while the rest of the framework (synthetic data, training engine,
kernels) is fully functional without them.

Storage format: same two-group layout the reference writes
(``images`` group of encoded/raw image arrays, ``masks`` group of stacked
(2, h, w) mask_miss/mask_all pairs, ``dataset`` attr JSON metadata list).
"""
from __future__ import annotations

import json

import numpy as np
import torch
from torch.utils.data import Dataset

from ..config import COCOSourceConfig
from .heatmapper import Heatmapper
from .transformer import Transformer, AugmentSelection

try:
    import h5py
    _H5_ERR = None
except Exception as e:  # pragma: no cover
    h5py = None
    _H5_ERR = e


class RawDataIterator:
    """Reads one record from the h5 file, augments it and generates GT
    (reference py_data_iterator.py:35-144)."""

    def __init__(self, global_config, source_config: COCOSourceConfig, shuffle=True,
                 augment=True, h5_file=None):
        """``h5_file``: optional pre-opened mapping with the same two-group
        layout (``dataset``/``images``/``masks`` of ``[()]``-indexable
        entries). Lets tests exercise the full read -> convert -> transform ->
        heatmap path with an in-memory fixture when h5py is unavailable."""
        if h5py is None and h5_file is None:
            raise RuntimeError("h5py is required for the COCO HDF5 data path") from _H5_ERR
        self.global_config = global_config
        self.source_config = source_config
        self.h5_path = source_config.source()
        self.h5 = h5_file  # else opened lazily per worker process
        self.keys = None
        self.shuffle = shuffle
        self.augment = augment
        self.heatmapper = Heatmapper(global_config)
        self.transformer = Transformer(global_config)
        if h5_file is not None:
            self._bind_groups()

    def _bind_groups(self):
        self.datum = self.h5["dataset"]
        self.images = self.h5["images"]
        self.masks = self.h5.get("masks")
        self.keys = list(self.datum.keys())

    def _ensure_open(self):
        if self.h5 is None:
            self.h5 = h5py.File(self.h5_path, "r")
            self._bind_groups()

    def num_keys(self):
        self._ensure_open()
        return len(self.keys)

    def read_data(self, key):
        entry = self.datum[key]
        meta = json.loads(entry[()])
        img = self.images[meta["image"]][()]
        if img.ndim == 2 or img.shape[-1] != 3:
            raise ValueError("expected decoded (H, W, 3) images in the h5 file")
        mask_pair = self.masks[meta["image"]][()] if self.masks is not None else None
        if mask_pair is not None:
            mask_miss, mask_all = mask_pair[0], mask_pair[1]
        else:
            mask_miss = np.ones(img.shape[:2], np.float32)
            mask_all = np.ones(img.shape[:2], np.float32)
        return img, mask_miss, mask_all, meta

    def gen(self, index):
        self._ensure_open()
        key = self.keys[index % len(self.keys)]
        img, mask_miss, mask_all, meta = self.read_data(key)
        meta = self.source_config.convert(meta, self.global_config)
        aug = (AugmentSelection.random(self.global_config.transform_params)
               if self.augment else AugmentSelection.unrandom())
        img, mask_miss, mask_all, meta = self.transformer.transform(
            img, mask_miss, mask_all, meta, aug=aug)
        labels = self.heatmapper.create_heatmaps(meta["joints"], mask_all)
        return (torch.from_numpy(np.ascontiguousarray(img)),
                torch.from_numpy(mask_miss[None].astype(np.float32)),
                torch.from_numpy(np.ascontiguousarray(labels)))


class MyDataset(Dataset):
    """torch Dataset adapter (reference data/mydataset.py:15-37)."""

    def __init__(self, global_config, config: COCOSourceConfig, shuffle=True,
                 augment=True, h5_file=None):
        self.iterator = RawDataIterator(global_config, config, shuffle, augment,
                                        h5_file=h5_file)

    def __len__(self):
        return self.iterator.num_keys()

    def __getitem__(self, index):
        return self.iterator.gen(index)


def build_coco_h5(ann_file: str, img_dir: str, out_path: str, image_size: int = 512):
    """Offline dataset builder (capability of reference data/coco_masks_hdf5.py):
    selects main persons (>=5 keypoints, area >= 32^2, not within 0.3 bbox of a
    previous main person), computes mask_miss / mask_all and writes the h5 file.
    Requires pycocotools + h5py (not present in this offline image)."""
    try:
        from pycocotools.coco import COCO
        from pycocotools import mask as maskUtils
    except Exception as e:  # pragma: no cover
        raise RuntimeError("pycocotools is required to build the COCO h5 dataset") from e
    if h5py is None:  # pragma: no cover
        raise RuntimeError("h5py is required to build the COCO h5 dataset") from _H5_ERR

    from PIL import Image

    coco = COCO(ann_file)
    ids = list(coco.imgs.keys())
    with h5py.File(out_path, "w") as h5:
        grp_data = h5.create_group("dataset")
        grp_img = h5.create_group("images")
        grp_mask = h5.create_group("masks")
        serial = 0
        for img_id in ids:
            anns = coco.loadAnns(coco.getAnnIds(imgIds=img_id))
            persons = [a for a in anns if a.get("num_keypoints", 0) > 0]
            if not persons:
                continue
            info = coco.loadImgs(img_id)[0]
            h, w = info["height"], info["width"]
            mask_all = np.zeros((h, w), np.float32)
            mask_miss = np.ones((h, w), np.float32)
            for a in anns:
                rle = maskUtils.frPyObjects(a["segmentation"], h, w)
                m = maskUtils.decode(rle)
                if m.ndim == 3:
                    m = m.max(axis=2)
                mask_all[m > 0] = 1
                if a.get("iscrowd", 0) or a.get("num_keypoints", 0) < 5 \
                        or a.get("area", 0) < 32 * 32:
                    mask_miss[m > 0] = 0
            prev_centers = []
            for a in persons:
                if a.get("num_keypoints", 0) < 5 or a.get("area", 0) < 32 * 32:
                    continue
                x, y, bw, bh = a["bbox"]
                center = (x + bw / 2, y + bh / 2)
                if any(abs(center[0] - cx) < 0.3 * bw and abs(center[1] - cy) < 0.3 * bh
                       for cx, cy in prev_centers):
                    continue
                prev_centers.append(center)
                joints = np.array(a["keypoints"], np.float32).reshape(-1, 3)
                # re-code visibility: coco v=2 visible->1, v=1 occluded->0, v=0 ->2
                v = joints[:, 2].copy()
                joints[:, 2] = np.where(v == 2, 1, np.where(v == 1, 0, 2))
                all_joints = []
                for b in persons:
                    j = np.array(b["keypoints"], np.float32).reshape(-1, 3)
                    vv = j[:, 2].copy()
                    j[:, 2] = np.where(vv == 2, 1, np.where(vv == 1, 0, 2))
                    all_joints.append(j.tolist())
                img_key = info["file_name"]
                if img_key not in grp_img:
                    arr = np.asarray(Image.open(f"{img_dir}/{img_key}").convert("RGB"))
                    grp_img.create_dataset(img_key, data=arr, compression="gzip")
                    grp_mask.create_dataset(
                        img_key, data=np.stack([mask_miss, mask_all]),
                        compression="gzip")
                meta = {"image": img_key, "objpos": list(center),
                        "scale_provided": bh / image_size, "joints": all_joints}
                grp_data.create_dataset(str(serial), data=json.dumps(meta))
                serial += 1
    return out_path
