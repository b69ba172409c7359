#!/bin/bash
# Stress config (BASELINE.json #5): 8-stage IMHN @768^2 — more steps than the
# round-1 3-step smoke, plus HBM occupancy evidence for the 288 GB sizing
# (VERDICT r1 weak #8). Batch sweep: per-GPU batch raised until HBM is
# meaningfully used; device-GT mode exercises the on-device pipeline too.
set -x
mkdir -p gpurun_out
for B in 4 16 32 48; do
  timeout 420 python bench.py --mode train --input 768 --nstack 8 --batch $B \
    --steps 6 --warmup 2 > gpurun_out/stress_b$B.log 2>&1
  echo "batch $B rc=$?:"
  tail -1 gpurun_out/stress_b$B.log
done
timeout 420 python bench.py --mode train --input 768 --nstack 8 --batch 16 \
  --data device-gt --steps 6 --warmup 2 > gpurun_out/stress_devgt.log 2>&1
echo "device-gt rc=$?:"
tail -1 gpurun_out/stress_devgt.log
rocm-smi --showmeminfo vram | tail -5
