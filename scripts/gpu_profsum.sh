mkdir -p gpurun_out
R=$GRAFT_REPO_ROOT
cd /tmp && export TMPDIR=/tmp
timeout 150 rocprofv3 --kernel-trace --stats -d /tmp/prof -o inf -- python $R/bench.py --mode infer --steps 10 --warmup 4 > /tmp/i.log 2>&1
timeout 150 rocprofv3 --kernel-trace --stats -d /tmp/prof -o tr -- python $R/bench.py --mode train --steps 4 --warmup 2 > /tmp/t.log 2>&1
python3 - "$R" <<'PYEOF'
import sqlite3, sys
R = sys.argv[1]
def dump(db, out, steps):
    con = sqlite3.connect(db)
    rows = list(con.execute("SELECT name, total_calls, total_duration, average, percentage FROM top_kernels"))
    tot = sum(r[2] for r in rows)
    with open(out, 'w') as f:
        f.write("# rocprofv3 --kernel-trace --stats on MI355X (end of round 1)\n")
        f.write(f"# total kernel time {tot/1e3:.1f} ms over {steps} steps = {tot/steps/1e3:.2f} ms/step GPU-busy\n")
        f.write(f"{'pct':>7} {'calls':>7} {'avg_us':>10}  kernel\n")
        for r in rows[:25]:
            f.write(f"{r[4]:6.2f}% {r[1]:7d} {r[3]:10.2f}  {r[0][:100]}\n")
dump('/tmp/prof/inf_results.db', R + '/gpurun_out/infer_kernels.txt', 14)
dump('/tmp/prof/tr_results.db', R + '/gpurun_out/train_step_kernels.txt', 6)
PYEOF
head -10 $R/gpurun_out/infer_kernels.txt
