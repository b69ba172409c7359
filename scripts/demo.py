#!/usr/bin/env python3
"""Single-image pose demo (reference demo_image.py equivalent).

Runs the full pipeline — ensemble forward, peak finding, limb scoring, greedy
person assembly — on one image and writes a skeleton overlay PNG.

    python scripts/demo.py --image path.jpg --ckpt checkpoints/PoseNet_52_epoch.pth
    python scripts/demo.py --synthetic          # no image / checkpoint needed

With --synthetic the demo renders GT-quality heatmaps for a random synthetic
scene through the same post-processing (useful offline: no weights exist in
this environment).
"""
from __future__ import annotations

import argparse
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from improved_body_parts_amd.config import GetConfig, InferenceParams, TrainingOpt  # noqa: E402
from improved_body_parts_amd.engine.inference import (  # noqa: E402
    find_connections, find_peaks, find_people, process, subsets_to_keypoints)
from improved_body_parts_amd.models import NetworkEval  # noqa: E402
from improved_body_parts_amd.utils.visualization import draw_people  # noqa: E402


def save_png(path, img):
    """Minimal PNG writer (no cv2/PIL guaranteed offline)."""
    try:
        from PIL import Image
        Image.fromarray(img).save(path)
        return
    except ImportError:
        pass
    import struct
    import zlib
    h, w = img.shape[:2]
    raw = b"".join(b"\x00" + img[i].tobytes() for i in range(h))

    def chunk(tag, data):
        c = tag + data
        return struct.pack(">I", len(data)) + c + struct.pack(">I", zlib.crc32(c))
    png = (b"\x89PNG\r\n\x1a\n"
           + chunk(b"IHDR", struct.pack(">IIBBBBB", w, h, 8, 2, 0, 0, 0))
           + chunk(b"IDAT", zlib.compress(raw))
           + chunk(b"IEND", b""))
    with open(path, "wb") as f:
        f.write(png)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--image", help="input image path")
    ap.add_argument("--ckpt", help="checkpoint (PoseNet_*_epoch.pth format)")
    ap.add_argument("--out", default="demo_out.png")
    ap.add_argument("--nstack", type=int, default=4)
    ap.add_argument("--synthetic", action="store_true",
                    help="run post-processing on synthetic GT maps (no model)")
    args = ap.parse_args()

    config = GetConfig("Canonical")
    params, model_params = InferenceParams().as_params_dict()

    if args.synthetic:
        from improved_body_parts_amd.data import Heatmapper, sample_people
        rng = np.random.default_rng(0)
        H = W = 512
        people = sample_people(rng, W, H, max_people=3)
        hm = Heatmapper(config)
        maps = hm.create_heatmaps(people, np.ones((H // 4, W // 4), np.float32))
        maps_t = torch.from_numpy(maps)
        up = torch.nn.functional.interpolate(maps_t[None], size=(H, W),
                                             mode="bicubic",
                                             align_corners=False)[0]
        heat = up[config.heat_start:].permute(1, 2, 0).contiguous()
        paf = up[:config.paf_layers].permute(1, 2, 0).contiguous()
        all_peaks = find_peaks(heat, params, config)
        conn, special = find_connections(all_peaks, paf, H, params, config)
        subset, candidate = find_people(conn, special, all_peaks, params, config)
        kps = subsets_to_keypoints(subset, candidate, config)
        image = np.full((H, W, 3), 32, np.uint8)
    else:
        if not args.image:
            ap.error("--image is required without --synthetic")
        from improved_body_parts_amd.engine.inference import _read_image
        image = _read_image(args.image)
        opt = TrainingOpt(nstack=args.nstack, batch_size=1)
        model = NetworkEval(opt, config, bn=True)
        if args.ckpt:
            ckpt = torch.load(args.ckpt, map_location="cpu", weights_only=False)
            model.load_state_dict(ckpt["weights"])
        if torch.cuda.is_available():
            model = model.cuda().bfloat16()
            for m in model.modules():
                if isinstance(m, torch.nn.modules.batchnorm._BatchNorm):
                    m.float()
        model.eval()
        kps = process(image[:, :, ::-1] if image.shape[2] == 3 else image,
                      model, config, params, model_params)

    print(f"found {len(kps)} people")
    canvas = draw_people(image, kps)
    save_png(args.out, canvas)
    print(f"wrote {args.out}")


if __name__ == "__main__":
    main()
