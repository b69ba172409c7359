#!/usr/bin/env python3
"""Training CLI — the role of the reference's four driver scripts
(train.py / train_parallel.py / train_distributed.py / train_distributed_SWA.py)
behind one entry point.

Single GPU (or CPU plumbing):
    python scripts/train.py --epochs 5 --batch 16

Multi-GPU, one process per GPU over RCCL/xGMI:
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 scripts/train.py --epochs 60 --batch 32

SWA fine-tune from a checkpoint (reference train_distributed_SWA.py):
    python scripts/train.py --swa --resume checkpoints/PoseNet_52_epoch.pth
"""
from __future__ import annotations

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from improved_body_parts_amd.config import GetConfig, TrainingOpt  # noqa: E402
from improved_body_parts_amd.data import SyntheticPoseDataset  # noqa: E402
from improved_body_parts_amd.engine import SWATrainer, Trainer  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", default="Canonical",
                    help="Canonical | Canonical384 | Canonical768 | DenseSkeleton")
    ap.add_argument("--nstack", type=int, default=4)
    ap.add_argument("--variant", default="imhn",
                    help="imhn | final | attention | light | independent | ae")
    ap.add_argument("--batch", type=int, default=16, help="per-GPU batch size")
    ap.add_argument("--epochs", type=int, default=60)
    ap.add_argument("--lr", type=float, default=2.5e-5, help="base LR per GPU")
    ap.add_argument("--resume", help="checkpoint path to resume from")
    ap.add_argument("--swa", action="store_true", help="SWA fine-tune mode")
    ap.add_argument("--ckpt-dir", default="checkpoints")
    ap.add_argument("--data", default="synthetic",
                    help="'synthetic' or a COCO h5 file (needs h5py)")
    ap.add_argument("--train-samples", type=int, default=2048,
                    help="synthetic dataset length per epoch")
    ap.add_argument("--num-workers", type=int, default=4)
    ap.add_argument("--no-sync-bn", action="store_true")
    ap.add_argument("--device-data", type=int, default=0, metavar="STEPS",
                    help="use the on-device GT generator for STEPS batches/epoch")
    args = ap.parse_args()

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if world_size > 1:
        import torch.distributed as dist
        dist.init_process_group(
            backend="nccl" if torch.cuda.is_available() else "gloo",
            init_method="env://")
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)

    config = GetConfig(args.config)
    opt = TrainingOpt(nstack=args.nstack, batch_size=args.batch,
                      learning_rate=args.lr, model_variant=args.variant,
                      nstack_weight=[1] * args.nstack)

    if args.data == "synthetic":
        train_ds = SyntheticPoseDataset(config, length=args.train_samples, seed=11)
        val_ds = SyntheticPoseDataset(config, length=max(args.batch * 4, 16),
                                      seed=101)
    else:
        from improved_body_parts_amd.config import COCOSourceConfig
        from improved_body_parts_amd.data import MyDataset
        train_ds = MyDataset(config, COCOSourceConfig(args.data),
                             shuffle=True, augment=True)
        val_ds = None

    cls = SWATrainer if args.swa else Trainer
    trainer = cls(opt, config, train_ds, val_ds, rank=rank,
                  local_rank=local_rank, world_size=world_size,
                  sync_bn=not args.no_sync_bn and world_size > 1,
                  num_workers=args.num_workers, checkpoint_dir=args.ckpt_dir,
                  device_synth_steps=args.device_data)
    if args.resume:
        trainer.resume(args.resume)
    trainer.fit(args.epochs)

    if world_size > 1:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
