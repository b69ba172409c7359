"""Browse a pose dataset (reference data/see_coco_data.py equivalent).

Works on either source:
  * a COCO-layout h5 file written by ``build_coco_h5`` (requires h5py), or
  * the offline synthetic generator (``--synthetic``, no dependencies).

Prints per-record stats and optionally dumps overlay PNGs of the image with
its GT keypoint/limb channels (the reference's matplotlib browse loop).

    python scripts/see_data.py --synthetic -n 4 --save-dir /tmp/browse
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np


def overlay(img, labels, config):
    """Image + max over keypoint channels upsampled, as a uint8 RGB array."""
    heat = labels[config.heat_start:config.bkg_start - 1].max(axis=0)
    heat = np.kron(heat, np.ones((config.stride, config.stride)))
    heat = heat[:img.shape[0], :img.shape[1]]
    out = (img * 255).astype(np.float32)
    out[..., 0] = np.clip(out[..., 0] + heat * 255, 0, 255)
    return out.astype(np.uint8)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--h5", default=None, help="coco h5 file to browse")
    ap.add_argument("--synthetic", action="store_true")
    ap.add_argument("--config", default="Canonical")
    ap.add_argument("-n", type=int, default=4)
    ap.add_argument("--save-dir", default=None)
    ap.add_argument("--bench", action="store_true",
                    help="augmentation/GT-generation throughput (reference "
                         "data/mydataset.py test_augmentation_speed; the "
                         "reference reports ~40 samples/s per CPU process)")
    args = ap.parse_args()

    from improved_body_parts_amd.config import COCOSourceConfig, GetConfig
    config = GetConfig(args.config)

    if args.synthetic or not args.h5:
        from improved_body_parts_amd.data import SyntheticPoseDataset
        ds = SyntheticPoseDataset(config, length=args.n, render=True, seed=7)
        get = lambda i: ds[i]
    else:
        from improved_body_parts_amd.data.coco import MyDataset
        ds = MyDataset(config, COCOSourceConfig(args.h5), augment=False)
        get = lambda i: ds[i]
        print(f"{len(ds)} records in {args.h5}")

    if args.bench:
        import time
        get(0)  # warm caches
        t0 = time.perf_counter()
        n = max(args.n, 16)
        for i in range(n):
            get(i)
        dt = time.perf_counter() - t0
        print(f"{n / dt:.1f} samples/s single process "
              f"(transform + GT generation; reference: ~40/s)")
        return

    for i in range(args.n):
        img, mask_miss, labels = (t.numpy() for t in get(i))
        kp = labels[config.heat_start:config.bkg_start - 1]
        paf = labels[:config.paf_layers]
        print(f"record {i}: image {img.shape} [{img.min():.2f},{img.max():.2f}]  "
              f"mask_miss zero-frac {(mask_miss < 0.5).mean():.3f}  "
              f"keypoint channels max {kp.max():.2f} ({int((kp.max(axis=(1, 2)) > 0.5).sum())}"
              f"/{kp.shape[0]} active)  paf max {paf.max():.2f}")
        if args.save_dir:
            os.makedirs(args.save_dir, exist_ok=True)
            from PIL import Image
            Image.fromarray(overlay(img, labels, config)).save(
                os.path.join(args.save_dir, f"record_{i}.png"))
            print(f"  wrote {args.save_dir}/record_{i}.png")


if __name__ == "__main__":
    main()
