#!/bin/bash
# PMC counters for the wgrad kernel (big 3x3 shape) and the fwd 1x1@128^2 —
# the two ratios furthest below MIOpen after the r2 rework.
set -x
R=$GRAFT_REPO_ROOT
mkdir -p $R/gpurun_out
cat > /tmp/wgrad_pmc_driver.py <<'PYEOF'
import sys, torch
sys.path.insert(0, __import__('os').environ['GRAFT_REPO_ROOT'])
from improved_body_parts_amd.ops import conv_kernels
CL = torch.channels_last
x = torch.randn(16, 256, 128, 128, device="cuda").bfloat16().contiguous(memory_format=CL)
dy = (torch.randn(16, 256, 128, 128, device="cuda") * 0.1).bfloat16().contiguous(memory_format=CL)
for _ in range(8):
    conv_kernels.conv_wgrad(x, dy, (256, 256, 3, 3), (1, 1), (1, 1), (1, 1))
x1 = torch.randn(16, 256, 128, 128, device="cuda").bfloat16().contiguous(memory_format=CL)
w1 = (torch.randn(128, 256, 1, 1, device="cuda") * 0.05).bfloat16()
for _ in range(8):
    conv_kernels.conv_fwd(x1, w1, (1, 1), (0, 0), (1, 1))
torch.cuda.synchronize()
PYEOF
cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_VALU_MFMA_BUSY_CYCLES SQ_WAIT_ANY SQ_ACTIVE_INST_ANY SQ_LDS_BANK_CONFLICT SQ_INSTS_VALU --output-format csv -d /tmp/wprof -o wpmc -- python /tmp/wgrad_pmc_driver.py > /tmp/wpmc.log 2>&1
echo "pmc=$?"
python - <<'PYEOF' > $R/gpurun_out/wgrad_pmc.txt 2>&1
import csv, glob, collections
f = sorted(glob.glob('/tmp/wprof/**/*counter_collection.csv', recursive=True))
print('files:', f)
agg = collections.defaultdict(lambda: collections.defaultdict(float))
disp = collections.Counter()
for fn in f:
    with open(fn) as fh:
        for row in csv.DictReader(fh):
            k = row.get('Kernel_Name', row.get('kernel_name', '?'))[:70]
            c = row.get('Counter_Name', row.get('counter_name', '?'))
            v = float(row.get('Counter_Value', row.get('counter_value', 0)))
            agg[k][c] += v
            disp[(k, c)] += 1
for k, d in sorted(agg.items(), key=lambda kv: -kv[1].get('SQ_WAVE_CYCLES', 0))[:8]:
    wc = d.get('SQ_WAVE_CYCLES', 1)
    print(k)
    print(f"   WAVE_CYC={wc:.3e} mfma={d.get('SQ_VALU_MFMA_BUSY_CYCLES',0)/wc*100:5.1f}% "
          f"wait={d.get('SQ_WAIT_ANY',0)/wc*100:5.1f}% active={d.get('SQ_ACTIVE_INST_ANY',0)/wc*100:5.1f}% "
          f"valu_insts={d.get('SQ_INSTS_VALU',0):.3e} lds_conf={d.get('SQ_LDS_BANK_CONFLICT',0):.3e}")
PYEOF
cat $R/gpurun_out/wgrad_pmc.txt

# second pass: memory counters (separate run — cannot combine with traces)
timeout 600 rocprofv3 --pmc FETCH_SIZE WRITE_SIZE SQ_WAVE_CYCLES SQ_VALU_MFMA_BUSY_CYCLES SQ_WAIT_ANY --output-format csv -d /tmp/wprof2 -o wpmc2 -- python /tmp/wgrad_pmc_driver.py > /tmp/wpmc2.log 2>&1
echo "pmc2=$?"
python - <<'PYEOF' >> $R/gpurun_out/wgrad_pmc.txt 2>&1
import csv, glob, collections
f = sorted(glob.glob('/tmp/wprof2/**/*counter_collection.csv', recursive=True))
print('=== memory counters ===')
agg = collections.defaultdict(lambda: collections.defaultdict(float))
for fn in f:
    with open(fn) as fh:
        for row in csv.DictReader(fh):
            k = row.get('Kernel_Name', row.get('kernel_name', '?'))[:70]
            c = row.get('Counter_Name', row.get('counter_name', '?'))
            agg[k][c] += float(row.get('Counter_Value', row.get('counter_value', 0)))
for k, d in sorted(agg.items(), key=lambda kv: -kv[1].get('SQ_WAVE_CYCLES', 0))[:6]:
    wc = d.get('SQ_WAVE_CYCLES', 1)
    print(k)
    print(f"   fetch={d.get('FETCH_SIZE',0)/1e6:.1f} MB write={d.get('WRITE_SIZE',0)/1e6:.1f} MB "
          f"mfma={d.get('SQ_VALU_MFMA_BUSY_CYCLES',0)/wc*100:5.1f}% wait={d.get('SQ_WAIT_ANY',0)/wc*100:5.1f}%")
PYEOF
tail -16 $R/gpurun_out/wgrad_pmc.txt
