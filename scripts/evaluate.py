#!/usr/bin/env python3
"""COCO evaluation CLI (the reference evaluate.py driver).

    python scripts/evaluate.py --ckpt checkpoints/PoseNet_52_epoch.pth \
        --ann data/annotations/person_keypoints_val2017.json \
        --images data/val2017 --max-images 500

Runs the device-resident ensemble forward + HIP post-processing over the
validation images and reports COCO AP via pycocotools (required, together
with an image reader — absent in the offline build image).
"""
from __future__ import annotations

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from improved_body_parts_amd.config import GetConfig, InferenceParams, TrainingOpt  # noqa: E402
from improved_body_parts_amd.engine.inference import validation  # noqa: E402
from improved_body_parts_amd.models import NetworkEval  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--ckpt", required=True)
    ap.add_argument("--ann", required=True, help="COCO person_keypoints json")
    ap.add_argument("--images", required=True, help="image directory")
    ap.add_argument("--config", default="Canonical")
    ap.add_argument("--nstack", type=int, default=4)
    ap.add_argument("--variant", default="imhn")
    ap.add_argument("--max-images", type=int, default=500,
                    help="evaluate the first N val images (reference used 500)")
    ap.add_argument("--scales", default="1.0",
                    help="comma-separated scale search, e.g. '0.8,1.0,1.2'")
    ap.add_argument("--dump-name", default="mi355_eval")
    args = ap.parse_args()

    config = GetConfig(args.config)
    opt = TrainingOpt(nstack=args.nstack, batch_size=1,
                      model_variant=args.variant,
                      nstack_weight=[1] * args.nstack)
    model = NetworkEval(opt, config, bn=True)
    ckpt = torch.load(args.ckpt, map_location="cpu", weights_only=False)
    weights = ckpt["weights"] if isinstance(ckpt, dict) and "weights" in ckpt else ckpt
    cleaned = {}
    for k, v in weights.items():
        k = k[len("module."):] if k.startswith("module.") else k
        k = k[len("posenet."):] if k.startswith("posenet.") else k
        if k.startswith("criterion."):
            continue
        cleaned[k] = v
    model.posenet.load_state_dict(cleaned)
    if torch.cuda.is_available():
        model = model.cuda().bfloat16()
        for m in model.modules():
            if isinstance(m, torch.nn.modules.batchnorm._BatchNorm):
                m.float()
    model.eval()

    params, model_params = InferenceParams().as_params_dict()
    params = dict(params)
    params["scale_search"] = [float(s) for s in args.scales.split(",")]

    from pycocotools.coco import COCO  # noqa: F401 — fail early with a clear error
    coco_ids = None
    if args.max_images:
        coco_ids = COCO(args.ann).getImgIds()[:args.max_images]
    validation(model, config, args.ann, args.images, args.dump_name,
               validation_ids=coco_ids, params=params,
               model_params=model_params)


if __name__ == "__main__":
    main()
