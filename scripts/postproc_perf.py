#!/usr/bin/env python3
"""End-to-end keypoint-assignment benchmark.

The reference's pure-Python peaks->connections->people stage runs at
5.2 images/s on a Xeon (reference README.md:68). Here the NMS + centroid
refinement + 20-point limb line integrals are HIP kernels and only the greedy
assembly stays on the host; this script measures the full assignment stage on
realistic multi-person maps at the reference's full-image resolution.

    python scripts/postproc_perf.py            # on a GPU box
    python scripts/postproc_perf.py --cpu      # host-only reference path
"""
from __future__ import annotations

import argparse
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from improved_body_parts_amd.config import GetConfig, InferenceParams  # noqa: E402
from improved_body_parts_amd.data import Heatmapper, sample_people  # noqa: E402
from improved_body_parts_amd.engine.inference import (  # noqa: E402
    find_connections, find_peaks, find_people, subsets_to_keypoints)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--cpu", action="store_true")
    ap.add_argument("--images", type=int, default=20)
    ap.add_argument("--size", type=int, default=512)
    ap.add_argument("--people", type=int, default=6)
    args = ap.parse_args()

    config = GetConfig("Canonical")
    params, _ = InferenceParams().as_params_dict()
    hm = Heatmapper(config)
    rng = np.random.default_rng(3)
    H = W = args.size

    device = "cpu" if args.cpu or not torch.cuda.is_available() else "cuda"
    scenes = []
    for _ in range(args.images):
        people = sample_people(rng, W, H, max_people=args.people)
        maps = hm.create_heatmaps(people, np.ones((H // 4, W // 4), np.float32))
        up = torch.nn.functional.interpolate(
            torch.from_numpy(maps)[None], size=(H, W), mode="bicubic",
            align_corners=False)[0]
        heat = up[config.heat_start:].permute(1, 2, 0).contiguous().to(device)
        paf = up[:config.paf_layers].permute(1, 2, 0).contiguous().to(device)
        scenes.append((heat, paf, len(people)))

    # warmup
    for heat, paf, _ in scenes[:3]:
        ap_ = find_peaks(heat, params, config)
        ca, sk = find_connections(ap_, paf, H, params, config)
        find_people(ca, sk, ap_, params, config)
    if device == "cuda":
        torch.cuda.synchronize()

    found = 0
    t0 = time.perf_counter()
    for heat, paf, n_gt in scenes:
        all_peaks = find_peaks(heat, params, config)
        conn, special = find_connections(all_peaks, paf, H, params, config)
        subset, cand = find_people(conn, special, all_peaks, params, config)
        kps = subsets_to_keypoints(subset, cand, config)
        found += len(kps)
    if device == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0

    fps = args.images / dt
    print(f"device={device} images={args.images} size={H} "
          f"people_found={found} assignment_fps={fps:.1f} "
          f"({dt / args.images * 1e3:.1f} ms/img) "
          f"vs_reference_5.2fps={fps / 5.2:.1f}x")


if __name__ == "__main__":
    main()
