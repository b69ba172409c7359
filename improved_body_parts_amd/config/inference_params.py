"""Inference post-processing parameters.

Capability parity with the reference's INI file + configobj reader
(reference utils/config:1-43 and utils/config_reader.py:1-41). Re-designed as a
plain dataclass with the same field names and defaults; ``config_reader()``
returns the same ``(params, model_params)`` pair of dicts the reference's
post-processing code consumes.
"""
from __future__ import annotations

from dataclasses import dataclass, field, asdict
from typing import List


@dataclass
class InferenceParams:
    # -- search / ensembling -------------------------------------------------
    use_gpu: int = 1
    GPUdeviceNumber: int = 0
    modelID: str = "1"
    octave: int = 3
    starting_range: float = 0.8
    ending_range: float = 2.0
    scale_search: List[float] = field(default_factory=lambda: [1.0])
    rotation_search: List[float] = field(default_factory=lambda: [0.0])
    # -- peak / connection thresholds ---------------------------------------
    thre1: float = 0.1          # keypoint peak threshold
    thre2: float = 0.1          # limb sample threshold
    thre3: float = 0.5
    mid_num: int = 20           # samples along each candidate limb segment
    connect_ration: float = 0.8  # fraction of samples that must clear thre2
    len_rate: float = 16.0      # limb length must be < len_rate * previous shorter limb
    connection_tole: float = 0.7
    offset_radius: int = 2      # sub-pixel centroid refinement radius
    min_num: int = 4
    # -- model geometry ------------------------------------------------------
    boxsize: int = 640
    stride: int = 4
    padValue: int = 128
    max_downsample: int = 64    # pad input to a multiple of this

    def as_params_dict(self):
        """Return the (params, model_params) dict pair of the reference reader."""
        d = asdict(self)
        model_keys = ("boxsize", "stride", "padValue", "max_downsample")
        model_params = {k: d.pop(k) for k in model_keys}
        return d, model_params


def config_reader(path: str | None = None):
    """Drop-in equivalent of reference utils/config_reader.py:config_reader().

    With no path, returns the built-in defaults; with a path, parses a simple
    ``key = value`` INI-style file (the reference's utils/config format) and
    overlays it.
    """
    p = InferenceParams()
    if path:
        section = None
        with open(path) as f:
            for line in f:
                line = line.split("#")[0].strip()
                if not line:
                    continue
                if line.startswith("["):
                    section = line.strip("[]")
                    continue
                if "=" not in line:
                    continue
                key, val = (x.strip() for x in line.split("=", 1))
                if not hasattr(p, key):
                    continue
                cur = getattr(p, key)
                if isinstance(cur, list):
                    setattr(p, key, [float(v) for v in val.strip("[]").split(",") if v.strip()])
                elif isinstance(cur, float):
                    setattr(p, key, float(val))
                elif isinstance(cur, int):
                    setattr(p, key, int(float(val)))
                else:
                    setattr(p, key, val)
    return p.as_params_dict()
