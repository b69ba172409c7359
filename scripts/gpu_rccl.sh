#!/bin/bash
# RCCL-on-hardware validation:
#  1. 2 ranks over NCCL(=RCCL) on the single leased MI355X — validates
#     process-group init, GradReducer bucket overlap and bench's multi-rank
#     path with real collectives. NCCL may refuse two ranks on one device
#     ("Duplicate GPU detected"); the log is kept either way.
#  2. Fallback/extra: world-size-1 RCCL init + an explicit all_reduce through
#     the GradReducer bucket path on device tensors.
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 420 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29631 bench.py --gpus 2 --mode train \
  --steps 8 --warmup 2 --batch 4 > gpurun_out/rccl_2rank.log 2>&1
echo "two_rank_rc=$?" | tee -a gpurun_out/rccl_2rank.log

timeout 300 python - > gpurun_out/rccl_1rank.log 2>&1 <<'PYEOF'
import os, time, torch, torch.distributed as dist
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29632")
dist.init_process_group("nccl", rank=0, world_size=1)
print("RCCL init ok:", dist.get_backend(), "world", dist.get_world_size())
x = torch.randn(1 << 22, device="cuda")
dist.all_reduce(x)          # real RCCL collective on device memory
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(20):
    dist.all_reduce(x)
torch.cuda.synchronize()
print("all_reduce 16MB x20: %.3f ms/call" % ((time.perf_counter() - t0) / 20 * 1e3))

# GradReducer bucket path on device tensors through the real backend
import sys
sys.path.insert(0, os.environ.get("GRAFT_REPO_ROOT", "/root/repo"))
from improved_body_parts_amd.parallel import GradReducer
model = torch.nn.Sequential(torch.nn.Linear(256, 256), torch.nn.ReLU(),
                            torch.nn.Linear(256, 64)).cuda().bfloat16()
red = GradReducer(model, bucket_cap_mb=0.05)
red.zero_grad()
y = model(torch.randn(32, 256, device="cuda").bfloat16()).float().pow(2).sum()
y.backward()
red.finalize()
torch.cuda.synchronize()
print("GradReducer over RCCL on device: ok, buckets =", len(red.buckets))
dist.destroy_process_group()
PYEOF
echo "one_rank_rc=$?"
tail -5 gpurun_out/rccl_2rank.log
tail -6 gpurun_out/rccl_1rank.log
