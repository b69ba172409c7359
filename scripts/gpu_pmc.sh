set -x
mkdir -p gpurun_out
R=$GRAFT_REPO_ROOT
timeout 420 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest=$?"
timeout 240 python bench.py --mode train --steps 15 --warmup 5 > gpurun_out/bench_train.log 2>&1
timeout 240 python bench.py --mode infer --steps 30 --warmup 10 > gpurun_out/bench_infer.log 2>&1
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY SQ_VALU_MFMA_BUSY_CYCLES --output-format csv -d /tmp/prof -o pmct -- python $R/bench.py --mode train --steps 3 --warmup 1 > /tmp/pmct.log 2>&1
echo "pmc=$?"
python - <<'PYEOF' > $R/gpurun_out/pmc_train.txt 2>&1
import csv, glob, collections
f = sorted(glob.glob('/tmp/prof/*counter_collection.csv'))
print('files:', f)
agg = collections.defaultdict(lambda: collections.defaultdict(float))
cnt = collections.Counter()
for fn in f:
    with open(fn) as fh:
        for row in csv.DictReader(fh):
            k = row.get('Kernel_Name', row.get('kernel_name', '?'))[:60]
            c = row.get('Counter_Name', row.get('counter_name', '?'))
            v = float(row.get('Counter_Value', row.get('counter_value', 0)))
            agg[k][c] += v
            cnt[k] += 1
rows = sorted(agg.items(), key=lambda kv: -kv[1].get('SQ_WAVE_CYCLES', 0))[:12]
for k, d in rows:
    wc = d.get('SQ_WAVE_CYCLES', 1)
    print(f"{k}")
    print(f"   WAVE_CYC={wc:.3e} mfma={d.get('SQ_VALU_MFMA_BUSY_CYCLES',0)/wc*100:5.1f}% "
          f"wait={d.get('SQ_WAIT_ANY',0)/wc*100:5.1f}% issue_stall={d.get('SQ_WAIT_INST_ANY',0)/wc*100:5.1f}% "
          f"active={d.get('SQ_ACTIVE_INST_ANY',0)/wc*100:5.1f}%")
PYEOF
cat $R/gpurun_out/pmc_train.txt | head -30
tail -2 $R/gpurun_out/bench_train.log; tail -2 $R/gpurun_out/bench_infer.log; tail -2 $R/gpurun_out/pytest_gpu.log
