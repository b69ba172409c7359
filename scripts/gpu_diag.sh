set -x
mkdir -p gpurun_out
R=$GRAFT_REPO_ROOT
cd /tmp && export TMPDIR=/tmp
timeout 240 rocprofv3 --kernel-trace --stats -d /tmp/prof -o bnperf -- python $R/scripts/bn_perf.py > /tmp/bnperf.log 2>&1
echo "bnperf=$?"
timeout 300 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY SQ_VALU_MFMA_BUSY_CYCLES --kernel-trace -d /tmp/prof -o pmci -- python $R/bench.py --mode infer --steps 5 --warmup 3 > /tmp/pmci.log 2>&1
echo "pmc=$?"
python - <<'PYEOF' > $R/gpurun_out/diag_summary.txt 2>&1
import sqlite3, glob
for db in sorted(glob.glob('/tmp/prof/*bnperf*.db')):
    con = sqlite3.connect(db)
    print('==', db)
    for r in con.execute("SELECT name, total_calls, total_duration, average, percentage FROM top_kernels LIMIT 12"):
        print(f"{r[4]:6.2f}%  {r[1]:6d} calls  avg {r[3]:9.2f}us  {r[0][:70]}")
for db in sorted(glob.glob('/tmp/prof/*pmci*.db')):
    con = sqlite3.connect(db)
    tabs = [t[0] for t in con.execute("SELECT name FROM sqlite_master WHERE type IN ('table','view')")]
    print('==', db)
    print([t for t in tabs if 'counter' in t.lower() or 'pmc' in t.lower()][:10])
    try:
        cols = [c[1] for c in con.execute("PRAGMA table_info(counters_collection)")]
        print('cols:', cols[:14])
    except Exception as e:
        print('err', e)
PYEOF
tail -20 /tmp/bnperf.log | head -12 >> $R/gpurun_out/diag_summary.txt
cat $R/gpurun_out/diag_summary.txt
