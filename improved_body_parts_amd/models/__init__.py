"""Model layer: IMHN stacked hourglass + loss.

Public API matches the reference's models/ package (posenet.py, layers_transposed.py,
loss_model.py) so user code and checkpoints carry over unchanged.
"""
from .layers import (Conv, Residual, BasicResidual, DilatedConv, Backbone,
                     Hourglass, SELayer)
from .posenet import PoseNet, Network, NetworkEval, Merge, Features
from .loss import MultiTaskLoss, MultiTaskLossParallel
from .variants import (PoseNetFinal, PoseNetAttention, PoseNetLight,
                       PoseNetIndependent, AEPoseNet, build_posenet, VARIANTS)

__all__ = [
    "Conv", "Residual", "BasicResidual", "DilatedConv", "Backbone", "Hourglass",
    "SELayer", "PoseNet", "Network", "NetworkEval", "Merge", "Features",
    "MultiTaskLoss", "MultiTaskLossParallel",
    "PoseNetFinal", "PoseNetAttention", "PoseNetLight", "PoseNetIndependent",
    "AEPoseNet", "build_posenet", "VARIANTS",
]
