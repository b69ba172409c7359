#!/usr/bin/env python3
"""torch.profiler attribution of the training step (CPU op -> GPU kernels).

Identifies which Python-level ops launch the long tail of small elementwise
kernels that rocprof shows but cannot attribute.
"""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from improved_body_parts_amd.config import GetConfig, TrainingOpt  # noqa: E402
from improved_body_parts_amd.data import SyntheticPoseDataset  # noqa: E402
from improved_body_parts_amd.engine import FusedSGD  # noqa: E402
from improved_body_parts_amd.models import Network  # noqa: E402


def main():
    config = GetConfig("Canonical")
    opt = TrainingOpt(nstack=4, batch_size=8, nstack_weight=[1] * 4)
    model = Network(opt, config, bn=True, dist=True).cuda().bfloat16()
    for m in model.modules():
        if isinstance(m, torch.nn.modules.batchnorm._BatchNorm):
            m.float()
    model.train()
    optimizer = FusedSGD(model.parameters(), lr=1e-5, momentum=0.9,
                         weight_decay=1e-4)
    ds = SyntheticPoseDataset(config, length=2, seed=5)
    img, mm, hm = ds[0]
    batch = tuple(t[None].expand(8, *([-1] * (t.dim()))).contiguous()
                  .cuda().bfloat16() for t in (img, mm, hm))

    def step():
        optimizer.zero_grad(set_to_none=True)
        loss = model(batch)
        loss.backward()
        optimizer.step()

    for _ in range(3):
        step()
    torch.cuda.synchronize()
    with torch.profiler.profile(
            activities=[torch.profiler.ProfilerActivity.CPU,
                        torch.profiler.ProfilerActivity.CUDA]) as prof:
        step()
        torch.cuda.synchronize()
    print(prof.key_averages().table(
        sort_by="self_cuda_time_total", row_limit=40, max_name_column_width=60))


if __name__ == "__main__":
    main()
