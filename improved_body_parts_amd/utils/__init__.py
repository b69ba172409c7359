from .util import (
    padRightDownCorner, center_pad, GaussianSmoothing, keypoint_heatmap_nms,
    refine_centroid, set_bn_eval, set_bn_eval_fp32, AverageMeter,
    adjust_learning_rate,
)

__all__ = [
    "padRightDownCorner", "center_pad", "GaussianSmoothing",
    "keypoint_heatmap_nms", "refine_centroid", "set_bn_eval",
    "set_bn_eval_fp32", "AverageMeter", "adjust_learning_rate",
]
