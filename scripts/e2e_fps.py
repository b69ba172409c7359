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

config = GetConfig("Canonical")
opt = TrainingOpt(nstack=4, batch_size=1)
model = NetworkEval(opt, config, bn=True).cuda().bfloat16()
for m in model.modules():
    if isinstance(m, torch.nn.modules.batchnorm._BatchNorm):
        m.float()
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
      f"({n_people} people found; 512^2, single scale, flip ensemble, "
      f"random-init weights)")
