"""End-to-end pipeline FPS: predict -> peaks -> connections -> greedy assembly
per image (the reference's whole-pipeline context: 7-8 FPS for the community
C++ rebuild, 5.2 FPS for the pure-Python assignment stage alone)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
from improved_body_parts_amd.config import GetConfig, TrainingOpt, InferenceParams
from improved_body_parts_amd.data import SyntheticPoseDataset
from improved_body_parts_amd.engine.inference import process
from improved_body_parts_amd.models import NetworkEval

import argparse
ap = argparse.ArgumentParser()
ap.add_argument("--train-steps", type=int, default=300,
                help="brief training first so the pipeline sees REAL peaks "
                     "(random-init nets produce none and make assembly free)")
args = ap.parse_args()

config = GetConfig("Canonical")
opt = TrainingOpt(nstack=4, batch_size=1)
model = NetworkEval(opt, config, bn=True).cuda().bfloat16()
for m in model.modules():
    if isinstance(m, torch.nn.modules.batchnorm._BatchNorm):
        m.float()
if args.train_steps:
    from improved_body_parts_amd.engine import FusedSGD
    from improved_body_parts_amd.models import Network
    topt = TrainingOpt(nstack=2, batch_size=8, nstack_weight=[1, 1])
    tnet = Network(topt, config, bn=True, dist=True).cuda().bfloat16()
    for m in tnet.modules():
        if isinstance(m, torch.nn.modules.batchnorm._BatchNorm):
            m.float()
    tnet.train()
    sgd = FusedSGD(tnet.parameters(), lr=4e-4, momentum=0.9, weight_decay=1e-4)
    tds = SyntheticPoseDataset(config, length=1 << 30, render=True, seed=11)
    for step in range(args.train_steps):
        b = [tds.generate(step * 8 + k)[:3] for k in range(8)]
        imgs_t = torch.stack([torch.from_numpy(x[0]) for x in b]).cuda().bfloat16()
        mms = torch.stack([torch.from_numpy(x[1]) for x in b]).cuda().bfloat16()
        hms = torch.stack([torch.from_numpy(np.ascontiguousarray(x[2])) for x in b]).cuda().bfloat16()
        lr = 4e-4 * min((step + 1) / 50.0, 1.0)
        for g in sgd.param_groups:
            g["lr"] = lr
        sgd.zero_grad(set_to_none=True)
        loss = tnet((imgs_t, mms, hms))
        if float(loss.detach()) < 1e6:
            loss.backward()
            sgd.step()
    # evaluate with the 2-stack trained net (weights are what matter for
    # realistic peak counts; note nstack in the printout)
    opt = topt
    model = NetworkEval(opt, config, bn=True).cuda().bfloat16()
    for m in model.modules():
        if isinstance(m, torch.nn.modules.batchnorm._BatchNorm):
            m.float()
    model.posenet.load_state_dict(tnet.posenet.state_dict())
model.eval()
p, mp = InferenceParams().as_params_dict()
ds = SyntheticPoseDataset(config, length=64, render=True, seed=3)
imgs = [ds.generate(i)[0] for i in range(24)]
for img in imgs[:4]:
    process(img, model, config, p, mp)          # warmup
torch.cuda.synchronize(); t0 = time.perf_counter()
n_people = 0
for img in imgs:
    n_people += len(process(img, model, config, p, mp))
torch.cuda.synchronize()
dt = time.perf_counter() - t0
print(f"end-to-end process() {len(imgs)/dt:.1f} img/s over {len(imgs)} images "
      f"({n_people} people found; 512^2, {opt.nstack}-stack, single scale, "
      f"flip ensemble, {'trained ' + str(args.train_steps) + ' steps' if args.train_steps else 'random-init'})")
