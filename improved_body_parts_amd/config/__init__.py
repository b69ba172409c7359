"""Configuration layer.

MI355X-native re-design of the reference's config system
(reference: config/config.py:8-266, utils/config + utils/config_reader.py).

Three mechanisms, mirroring the reference's capabilities:
  * ``TrainingOpt``          - training hyper-parameters (reference config/config.py:8-23)
  * ``CanonicalConfig``      - skeleton / channel-layout definition (reference config/config.py:52-134)
  * ``GetConfig(name)``      - registry lookup (reference config/config.py:239-260)
  * ``COCOSourceConfig``     - COCO joint-order -> canonical adapter (reference config/config.py:137-233)
  * ``InferenceParams``      - post-process knobs (reference utils/config INI, utils/config_reader.py)

Unlike the reference (four whole-file config forks: config2/config_dense/config_final),
variants here are parameterised constructors registered in ``Configs``.
"""
from .canonical import (
    TrainingOpt,
    TransformationParams,
    CanonicalConfig,
    DenseSkeletonConfig,
    COCOSourceConfig,
    Configs,
    GetConfig,
)
from .inference_params import InferenceParams, config_reader

__all__ = [
    "TrainingOpt",
    "TransformationParams",
    "CanonicalConfig",
    "DenseSkeletonConfig",
    "COCOSourceConfig",
    "Configs",
    "GetConfig",
    "InferenceParams",
    "config_reader",
]
