"""Training engine.

Capability parity with the reference's four drivers (train.py,
train_parallel.py, train_distributed.py, train_distributed_SWA.py), re-designed
as ONE engine with modes instead of four forked scripts:

  * single-GPU and multi-GPU one-process-per-GPU (RCCL over xGMI) — the
    reference's nn.DataParallel path is deliberately not carried over
    (DDP-style is the only multi-GPU path worth building).
  * native bf16 compute with fp32 master weights in the fused SGD
    (replaces Apex amp O1 + loss scaling; bf16 needs no scaler).
  * bucketed gradient all-reduce overlapped with backward (GradReducer)
    instead of Apex's delay_allreduce flat reduce.
  * checkpoint format preserved EXACTLY: ``{'weights': state_dict (un-prefixed),
    'optimizer_weight', 'train_loss', 'epoch'}`` -> ``PoseNet_{epoch}_epoch.pth``
    (reference train.py:151-162, train_distributed.py:304-324).
  * per-iteration LR schedule with 3-epoch warm-up + step decay
    (reference train_distributed.py:382-400), loss-explosion batch dropping
    (reference :259-261), rank-0 logging to ``checkpoints/log``.
  * SWA fine-tune mode (reference train_distributed_SWA.py): cyclic LR +
    weight averaging every ``swa_freq`` epochs, BN frozen.
"""
from __future__ import annotations

import os
import time

import torch
from torch.utils.data import DataLoader
from torch.utils.data.distributed import DistributedSampler

from ..models import Network
from ..parallel import GradReducer, convert_syncbn, reduce_tensor
from ..utils import AverageMeter, adjust_learning_rate
from ..engine.optimizer import FusedSGD


def save_checkpoint(model, optimizer, train_loss, epoch, directory="checkpoints"):
    """Write the reference-format checkpoint (train_distributed.py:312-324)."""
    os.makedirs(directory, exist_ok=True)
    module = model.module if hasattr(model, "module") else model
    state = {
        "weights": {k: v.float() if v.is_floating_point() else v
                    for k, v in module.state_dict().items()},
        "optimizer_weight": optimizer.state_dict(),
        "train_loss": float(train_loss),
        "epoch": int(epoch),
    }
    path = os.path.join(directory, f"PoseNet_{epoch}_epoch.pth")
    torch.save(state, path)
    return path


def load_checkpoint(model, path, optimizer=None, device="cpu", strict=True):
    """Resume from a reference-format checkpoint (reference train.py:61-81,
    train_distributed.py:149-197); accepts checkpoints written by the reference
    itself. Returns (epoch, train_loss)."""
    ckpt = torch.load(path, map_location="cpu", weights_only=False)
    weights = ckpt["weights"]
    module = model.module if hasattr(model, "module") else model
    # the reference sometimes saves with a 'module.' prefix — strip it
    cleaned = {(k[len("module."):] if k.startswith("module.") else k): v
               for k, v in weights.items()}
    missing, unexpected = module.load_state_dict(cleaned, strict=False)
    if strict and (missing or unexpected):
        raise RuntimeError(f"checkpoint mismatch: missing={missing[:5]} "
                           f"unexpected={unexpected[:5]}")
    if optimizer is not None and ckpt.get("optimizer_weight"):
        try:
            optimizer.load_state_dict(ckpt["optimizer_weight"])
            for state in optimizer.state.values():
                for k, v in state.items():
                    if torch.is_tensor(v):
                        state[k] = v.to(device)
        except Exception:
            pass  # optimizer layout changed (e.g. fused vs plain) — weights still loaded
    return ckpt.get("epoch", 0), ckpt.get("train_loss", float("inf"))


class Trainer:
    def __init__(self, opt, config, train_dataset, val_dataset=None, *,
                 rank=0, local_rank=0, world_size=1, use_bn=True, sync_bn=None,
                 num_workers=2, checkpoint_dir="checkpoints", log_file=None,
                 device=None, device_synth_steps=0):
        self.opt = opt
        self.config = config
        self.rank = rank
        self.world_size = world_size
        self.checkpoint_dir = checkpoint_dir
        self.log_file = log_file or os.path.join(checkpoint_dir, "log")
        self.device = device or (
            torch.device("cuda", local_rank) if torch.cuda.is_available()
            else torch.device("cpu"))
        self.bf16 = (opt.dtype == "bf16" and self.device.type == "cuda")

        model = Network(opt, config, bn=use_bn, dist=True)
        if sync_bn is None:
            sync_bn = world_size > 1
        if sync_bn and world_size > 1:
            model = convert_syncbn(model)
        self.model = model.to(self.device)
        if self.bf16:
            # bf16 activations/weights; BN keeps fp32 stats inside the kernels
            self.model = self.model.bfloat16()
            for m in self.model.modules():
                if isinstance(m, torch.nn.modules.batchnorm._BatchNorm):
                    m.float()

        base_lr = opt.learning_rate * world_size  # reference train_distributed.py:123-124
        self.base_lr = base_lr
        self.optimizer = FusedSGD(self.model.parameters(), lr=base_lr,
                                  momentum=opt.momentum,
                                  weight_decay=opt.weight_decay)
        self.reducer = (GradReducer(self.model) if world_size > 1 else None)

        if device_synth_steps and self.device.type == "cuda":
            # GPU-resident data: GT heatmaps from the HIP batched generator,
            # different stream per rank (north-star: 512^2 pipeline on device)
            from ..data import DeviceGTSyntheticLoader
            self.train_sampler = None
            self.train_loader = DeviceGTSyntheticLoader(
                config,
                batch_size=opt.batch_size, steps_per_epoch=device_synth_steps,
                seed=rank + 1, device=self.device,
                dtype=torch.bfloat16 if self.bf16 else torch.float32)
        else:
            self.train_sampler = (DistributedSampler(train_dataset)
                                  if world_size > 1 else None)
            self.train_loader = DataLoader(
                train_dataset, batch_size=opt.batch_size,
                shuffle=self.train_sampler is None, sampler=self.train_sampler,
                num_workers=num_workers, pin_memory=self.device.type == "cuda",
                drop_last=True)
        self.val_loader = (DataLoader(val_dataset, batch_size=opt.batch_size,
                                      shuffle=False, num_workers=num_workers,
                                      pin_memory=self.device.type == "cuda")
                           if val_dataset is not None else None)
        self.best_loss = float("inf")
        self.start_epoch = 0

    # ------------------------------------------------------------------ public
    def resume(self, path):
        epoch, loss = load_checkpoint(self.model, path, self.optimizer,
                                      device=self.device, strict=False)
        self.start_epoch = epoch + 1
        self.best_loss = loss
        return epoch

    def fit(self, epochs):
        for epoch in range(self.start_epoch, epochs):
            train_loss = self.train_epoch(epoch)
            val_loss = self.evaluate() if self.val_loader is not None else None
            if self.rank == 0:
                self._log(f"epoch {epoch} train_loss {train_loss:.6f}"
                          + (f" val_loss {val_loss:.6f}" if val_loss is not None else ""))
                save_checkpoint(self.model, self.optimizer, train_loss, epoch,
                                self.checkpoint_dir)
        return self.best_loss

    def train_epoch(self, epoch, max_iters=None):
        self.model.train()
        if self.train_sampler is not None:
            self.train_sampler.set_epoch(epoch)
        elif hasattr(self.train_loader, "set_epoch"):
            self.train_loader.set_epoch(epoch)  # device-GT stream
        meter = AverageMeter()
        iters_per_epoch = len(self.train_loader)
        t0 = time.time()
        for it, batch in enumerate(self.train_loader):
            if max_iters is not None and it >= max_iters:
                break
            adjust_learning_rate(self.optimizer, epoch, it, iters_per_epoch,
                                 self.base_lr, self.opt.warmup_epochs,
                                 self.opt.lr_decay_every, self.opt.lr_decay_factor)
            loss = self.train_step(batch)
            if loss is None:
                continue
            meter.update(loss)
            if self.rank == 0 and it % 10 == 0:
                self._log(f"epoch {epoch} iter {it}/{iters_per_epoch} "
                          f"loss {meter.avg:.6f} "
                          f"({(it + 1) * self.opt.batch_size * self.world_size / (time.time() - t0):.1f} img/s)")
        if self.world_size > 1:
            avg = reduce_tensor(torch.tensor(meter.avg, device=self.device))
            return float(avg)
        return meter.avg

    def train_step(self, batch):
        """One optimisation step. Returns the loss value, or None if the batch
        was dropped by the loss-explosion guard."""
        images, mask_miss, heatmaps = (t.to(self.device, non_blocking=True)
                                       for t in batch)
        if self.bf16:
            images = images.bfloat16()
            mask_miss = mask_miss.bfloat16()
            heatmaps = heatmaps.bfloat16()
        if self.reducer is not None:
            self.reducer.zero_grad()
        else:
            self.optimizer.zero_grad(set_to_none=True)
        loss = self.model((images, mask_miss, heatmaps))
        lv = float(loss.detach())
        if lv > self.opt.loss_explosion_thre or lv != lv:
            # drop exploding batches (reference train_distributed.py:259-261)
            self._log(f"dropping batch with loss {lv}")
            return None
        loss.backward()
        if self.reducer is not None:
            self.reducer.finalize()
        self.optimizer.step()
        return lv

    @torch.no_grad()
    def evaluate(self):
        self.model.eval()
        meter = AverageMeter()
        for batch in self.val_loader:
            images, mask_miss, heatmaps = (t.to(self.device, non_blocking=True)
                                           for t in batch)
            if self.bf16:
                images = images.bfloat16()
                mask_miss = mask_miss.bfloat16()
                heatmaps = heatmaps.bfloat16()
            _, loss = self.model((images, mask_miss, heatmaps))
            meter.update(float(loss))
        self.model.train()
        return meter.avg

    # ------------------------------------------------------------------- misc
    def _log(self, msg):
        if self.rank != 0:
            return
        line = f"[{time.strftime('%H:%M:%S')}] {msg}"
        print(line, flush=True)
        try:
            os.makedirs(os.path.dirname(self.log_file) or ".", exist_ok=True)
            with open(self.log_file, "a") as f:
                f.write(line + "\n")
        except OSError:
            pass


class SWATrainer(Trainer):
    """Stochastic Weight Averaging fine-tune (reference train_distributed_SWA.py):
    cyclic LR between lr_max and lr_min over ``swa_freq`` epochs; running average
    of weights updated at each cycle end; BatchNorm layers frozen."""

    def __init__(self, *args, swa_freq=5, lr_max=1e-5, lr_min=1e-6, **kwargs):
        super().__init__(*args, **kwargs)
        self.swa_freq = swa_freq
        self.lr_max = lr_max
        self.lr_min = lr_min
        self.swa_state = None
        self.swa_count = 0
        for m in self.model.modules():  # freeze BN (reference :219-221)
            if isinstance(m, torch.nn.modules.batchnorm._BatchNorm):
                m.eval()

    def _cyclic_lr(self, epoch, it, iters_per_epoch):
        t = ((epoch % self.swa_freq) + it / max(iters_per_epoch, 1)) / self.swa_freq
        return (1 - t) * self.lr_max + t * self.lr_min

    def train_epoch(self, epoch, max_iters=None):
        self.model.train()
        for m in self.model.modules():
            if isinstance(m, torch.nn.modules.batchnorm._BatchNorm):
                m.eval()
        if self.train_sampler is not None:
            self.train_sampler.set_epoch(epoch)
        elif hasattr(self.train_loader, "set_epoch"):
            self.train_loader.set_epoch(epoch)  # device-GT stream
        meter = AverageMeter()
        iters_per_epoch = len(self.train_loader)
        for it, batch in enumerate(self.train_loader):
            if max_iters is not None and it >= max_iters:
                break
            lr = self._cyclic_lr(epoch, it, iters_per_epoch)
            for g in self.optimizer.param_groups:
                g["lr"] = lr
            loss = self.train_step(batch)
            if loss is not None:
                meter.update(loss)
        if (epoch + 1) % self.swa_freq == 0:
            self.update_swa()
        return meter.avg

    def update_swa(self):
        module = self.model.module if hasattr(self.model, "module") else self.model
        sd = {k: v.detach().float().clone() for k, v in module.state_dict().items()
              if v.is_floating_point()}
        if self.swa_state is None:
            self.swa_state = sd
        else:
            n = self.swa_count
            for k in self.swa_state:
                self.swa_state[k].mul_(n / (n + 1)).add_(sd[k], alpha=1 / (n + 1))
        self.swa_count += 1

    def swap_swa_weights(self):
        """Load the averaged weights into the model (optimizer.swap_swa_sgd)."""
        if self.swa_state is None:
            return
        module = self.model.module if hasattr(self.model, "module") else self.model
        own = module.state_dict()
        for k, v in self.swa_state.items():
            own[k].copy_(v.to(own[k].dtype))
