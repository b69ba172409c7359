#!/usr/bin/env python3
"""Find the FIRST module (in backward execution order) whose grad_output is
sane but grad_input explodes, on the MFMA path."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from improved_body_parts_amd.config import GetConfig, TrainingOpt
from improved_body_parts_amd.data import DeviceGTSyntheticLoader
from improved_body_parts_amd.models import Network

config = GetConfig("Canonical")
opt = TrainingOpt(nstack=4, batch_size=8, nstack_weight=[1]*4)
torch.manual_seed(7)
model = Network(opt, config, bn=True, dist=True).cuda().bfloat16()
for m in model.modules():
    if isinstance(m, torch.nn.modules.batchnorm._BatchNorm):
        m.float()
model.train()
loader = DeviceGTSyntheticLoader(config, 8, steps_per_epoch=1, seed=1,
                                 dtype=torch.bfloat16)
batch = next(iter(loader))

events = []
def mk(name):
    def hook(mod, gin, gout):
        go = max((float(g.float().abs().max()) for g in gout
                  if torch.is_tensor(g)), default=0.0)
        gi = max((float(g.float().abs().max()) for g in gin
                  if torch.is_tensor(g)), default=0.0)
        events.append((name, type(mod).__name__, go, gi))
    return hook

from improved_body_parts_amd.models.layers import Conv, Residual, SELayer
for n, m in model.named_modules():
    if isinstance(m, (Conv, Residual, SELayer)):
        m.register_full_backward_hook(mk(n))

loss = model(batch)
loss.backward()
torch.cuda.synchronize()
print(f"loss {float(loss):.2f}, {len(events)} backward events")
# first event (in execution order) where grad blows up
prev_ok = None
for i, (n, t, go, gi) in enumerate(events):
    if go > 1e3 or gi > 1e3:
        print(f"FIRST explosion at event {i}: {n} ({t}) grad_out_max={go:.3e} grad_in_max={gi:.3e}")
        for j in range(max(0, i - 4), min(len(events), i + 4)):
            n2, t2, go2, gi2 = events[j]
            print(f"  [{j}] {n2} ({t2}) out={go2:.3e} in={gi2:.3e}")
        break
else:
    print("no explosion; max grad:", max(max(go, gi) for _, _, go, gi in events))
