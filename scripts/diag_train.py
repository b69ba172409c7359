#!/usr/bin/env python3
"""Per-iteration loss trace @512^2 to distinguish divergence from guard noise."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from improved_body_parts_amd.config import GetConfig, TrainingOpt
from improved_body_parts_amd.data import DeviceGTSyntheticLoader
from improved_body_parts_amd.engine import FusedSGD
from improved_body_parts_amd.models import Network
from improved_body_parts_amd.utils import adjust_learning_rate

config = GetConfig("Canonical")
opt = TrainingOpt(nstack=4, batch_size=16, nstack_weight=[1]*4)
model = Network(opt, config, bn=True, dist=True).cuda().bfloat16()
for m in model.modules():
    if isinstance(m, torch.nn.modules.batchnorm._BatchNorm):
        m.float()
model.train()
sgd = FusedSGD(model.parameters(), lr=opt.learning_rate, momentum=0.9,
               weight_decay=opt.weight_decay)
loader = DeviceGTSyntheticLoader(config, 16, steps_per_epoch=40, seed=1,
                                 dtype=torch.bfloat16)
for it, (img, mm, hm) in enumerate(loader):
    lr = adjust_learning_rate(sgd, 0, it, 40, opt.learning_rate)
    sgd.zero_grad(set_to_none=True)
    loss = model((img, mm, hm))
    loss.backward()
    sgd.step()
    print(f"it {it:3d} lr {lr:.2e} loss {float(loss):.1f}")
# eval-mode loss on one fresh batch
model.eval()
loader2 = DeviceGTSyntheticLoader(config, 16, steps_per_epoch=1, seed=9,
                                  dtype=torch.bfloat16)
with torch.no_grad():
    for img, mm, hm in loader2:
        out, l = model((img, mm, hm))
        print("eval loss", float(l))
