set -x
mkdir -p gpurun_out
R=$GRAFT_REPO_ROOT
timeout 300 python scripts/conv_perf.py > gpurun_out/conv_perf.txt 2>&1
timeout 420 python -m pytest tests/test_ops_gpu.py -q > gpurun_out/pytest_ops.log 2>&1
echo "pytest=$?"
timeout 240 python bench.py --mode train --steps 15 --warmup 5 > gpurun_out/bench_train.log 2>&1
timeout 240 python bench.py --mode infer --steps 30 --warmup 10 > gpurun_out/bench_infer.log 2>&1
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats -d /tmp/prof -o tr -- python $R/bench.py --mode train --steps 5 --warmup 2 > /tmp/tr.log 2>&1
timeout 300 rocprofv3 --kernel-trace --stats -d /tmp/prof -o inf -- python $R/bench.py --mode infer --steps 10 --warmup 5 > /tmp/inf.log 2>&1
python3 - "$R" <<'PYEOF'
import sqlite3, sys, glob
R = sys.argv[1]
def dump(db, out, steps):
    con = sqlite3.connect(db)
    rows = list(con.execute("SELECT name, total_calls, total_duration, average, percentage FROM top_kernels"))
    tot = sum(r[2] for r in rows)
    with open(out, 'w') as f:
        f.write(f"# rocprofv3 --kernel-trace --stats on MI355X\n")
        f.write(f"# total kernel time {tot/1e3:.1f} ms over {steps} steps = {tot/steps/1e3:.2f} ms/step GPU-busy\n")
        f.write(f"{'pct':>7} {'calls':>7} {'avg_us':>10}  kernel\n")
        for r in rows[:25]:
            f.write(f"{r[4]:6.2f}% {r[1]:7d} {r[3]:10.2f}  {r[0][:100]}\n")
dump('/tmp/prof/tr_results.db', R + '/gpurun_out/train_step_kernels.txt', 7)
dump('/tmp/prof/inf_results.db', R + '/gpurun_out/infer_kernels.txt', 15)
PYEOF
tail -2 $R/gpurun_out/bench_train.log; tail -2 $R/gpurun_out/bench_infer.log; tail -2 $R/gpurun_out/pytest_ops.log; cat $R/gpurun_out/conv_perf.txt | tail -10
