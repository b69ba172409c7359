import os, sys; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
"""Count conv FLOPs of the 4-stage IMHN @512^2 per image."""
import torch
from improved_body_parts_amd.config import GetConfig, TrainingOpt
from improved_body_parts_amd.models import PoseNet

cfg = GetConfig("Canonical")
opt = TrainingOpt()
net = PoseNet(opt.nstack, opt.hourglass_inp_dim, cfg.num_layers, bn=True,
              increase=opt.increase, init_weights=False)
flops = {}
hooks = []
def hook(mod, inp, out):
    n, c, h, w = out.shape
    k = mod.kernel_size[0] * mod.kernel_size[1]
    f = 2 * n * h * w * c * mod.in_channels * k
    key = (mod.in_channels, c, h, mod.kernel_size[0], mod.dilation[0])
    flops[key] = (flops.get(key, (0, 0))[0] + f, flops.get(key, (0, 0))[1] + 1)
for m in net.modules():
    if isinstance(m, torch.nn.Conv2d):
        hooks.append(m.register_forward_hook(hook))
with torch.no_grad():
    net(torch.rand(1, 512, 512, 3))
tot = sum(v[0] for v in flops.values())
print(f"total conv GFLOP/image: {tot/1e9:.1f}")
for key, (f, cnt) in sorted(flops.items(), key=lambda kv: -kv[1][0])[:18]:
    print(f"  cin{key[0]:4d} cout{key[1]:4d} hw{key[2]:4d} k{key[3]} d{key[4]}: "
          f"{f/1e9:7.2f} GF ({100*f/tot:4.1f}%) x{cnt}")
