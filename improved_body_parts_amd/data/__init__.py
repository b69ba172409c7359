"""Data layer: GT generation, augmentation, synthetic + COCO datasets.

Mirrors the capabilities of the reference's data/ + py_cocodata_server/ packages
(SURVEY.md §2 L1/L2), with the heatmapper doubled as a HIP kernel on device.
"""
from .heatmapper import Heatmapper, create_heatmaps_device, limb_gaussian
from .transformer import Transformer, AugmentSelection
from .synthetic import SyntheticPoseDataset, DeviceGTSyntheticLoader, sample_people
from .coco import MyDataset, RawDataIterator, build_coco_h5

__all__ = [
    "Heatmapper", "create_heatmaps_device", "limb_gaussian",
    "Transformer", "AugmentSelection",
    "SyntheticPoseDataset", "DeviceGTSyntheticLoader", "sample_people",
    "MyDataset", "RawDataIterator", "build_coco_h5",
]
