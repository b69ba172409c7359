#!/usr/bin/env python3
"""Localize the first non-finite tensor in the 512^2 training loop."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from improved_body_parts_amd.config import GetConfig, TrainingOpt
from improved_body_parts_amd.data import DeviceGTSyntheticLoader
from improved_body_parts_amd.engine import FusedSGD
from improved_body_parts_amd.models import Network

config = GetConfig("Canonical")
opt = TrainingOpt(nstack=4, batch_size=16, nstack_weight=[1]*4)
model = Network(opt, config, bn=True, dist=True).cuda().bfloat16()
for m in model.modules():
    if isinstance(m, torch.nn.modules.batchnorm._BatchNorm):
        m.float()
model.train()
sgd = FusedSGD(model.parameters(), lr=2.5e-5, momentum=0.9, weight_decay=2e-4)
loader = DeviceGTSyntheticLoader(config, 16, steps_per_epoch=30, seed=1,
                                 dtype=torch.bfloat16)

def check_params(tag):
    for n, p in model.named_parameters():
        if not torch.isfinite(p).all():
            print(f"[{tag}] PARAM non-finite: {n}")
            return n
    for n, b in model.named_buffers():
        if b.is_floating_point() and not torch.isfinite(b).all():
            print(f"[{tag}] BUFFER non-finite: {n}")
            return n
    return None

bad_batch = None
for it, batch in enumerate(loader):
    sgd.zero_grad(set_to_none=True)
    loss = model(batch)
    lv = float(loss)
    print(f"it {it:3d} loss {lv:.1f}", flush=True)
    if lv != lv:
        bad_batch = batch
        print("loss NaN at iter", it)
        break
    loss.backward()
    # check grads for non-finite
    for n, p in model.named_parameters():
        if p.grad is not None and not torch.isfinite(p.grad.float()).all():
            print(f"  GRAD non-finite after bwd: {n}")
            bad_batch = batch
            break
    if bad_batch is not None:
        break
    sgd.step()
    if check_params(f"after step {it}"):
        bad_batch = batch
        break

if bad_batch is not None:
    print("== rerunning forward with hooks ==")
    hooks = []
    first = []
    def mk(name):
        def h(mod, inp, out):
            if first: return
            outs = out if isinstance(out, (list, tuple)) else [out]
            for o in outs:
                if torch.is_tensor(o) and o.is_floating_point():
                    if not torch.isfinite(o.float()).all():
                        first.append(name)
                        print("FIRST non-finite output at:", name,
                              type(mod).__name__)
                        return
        return h
    for n, m in model.named_modules():
        hooks.append(m.register_forward_hook(mk(n)))
    with torch.no_grad():
        model(bad_batch)
    if not first:
        print("forward clean on re-run (non-deterministic or backward-side)")
else:
    print("NO NaN in 30 iters")
