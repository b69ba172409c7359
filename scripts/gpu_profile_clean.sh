#!/bin/bash
# Warmup-free steady-state kernel tables (the round-1
# tables were contaminated by MIOpen find-mode solver evaluations).
# Strategy: trace MANY steps, then keep only dispatches from the last 50% of
# the trace window and aggregate per kernel.
set -x
R=$GRAFT_REPO_ROOT
mkdir -p $R/gpurun_out
cd /tmp && export TMPDIR=/tmp
timeout 420 rocprofv3 --kernel-trace --output-format csv -d /tmp/cprof -o inf -- \
  python $R/bench.py --mode infer --steps 60 --warmup 20 > /tmp/ci.log 2>&1
echo "infer=$?"; tail -1 /tmp/ci.log
timeout 420 rocprofv3 --kernel-trace --output-format csv -d /tmp/cprof -o tr -- \
  python $R/bench.py --mode train --steps 12 --warmup 5 > /tmp/ct.log 2>&1
echo "train=$?"; tail -1 /tmp/ct.log
python3 - "$R" <<'PYEOF'
import csv, glob, sys, collections
R = sys.argv[1]

def clean_table(tag, out, steps_total, steps_timed, frac=0.5):
    files = glob.glob(f'/tmp/cprof/**/{tag}*kernel_trace.csv', recursive=True)
    rows = []
    for fn in files:
        with open(fn) as fh:
            for r in csv.DictReader(fh):
                try:
                    s = int(r.get('Start_Timestamp') or r.get('start_timestamp'))
                    e = int(r.get('End_Timestamp') or r.get('end_timestamp'))
                except (TypeError, ValueError):
                    continue
                rows.append((r.get('Kernel_Name') or r.get('kernel_name', '?'), s, e))
    if not rows:
        print(f'{tag}: no dispatches found in {files}')
        return
    t0 = min(r[1] for r in rows)
    t1 = max(r[2] for r in rows)
    cut = t0 + (t1 - t0) * frac
    agg = collections.defaultdict(lambda: [0, 0.0])
    kept_span = (t1 - cut) / 1e9
    for name, s, e in rows:
        if s < cut:
            continue
        agg[name][0] += 1
        agg[name][1] += (e - s) / 1e3  # us
    tot = sum(v[1] for v in agg.values())
    est_steps = steps_timed * kept_span and None
    with open(out, 'w') as f:
        f.write(f'# steady-state kernel table (last {int((1-frac)*100)}% of a '
                f'{steps_total}-step trace window; warmup excluded)\n')
        f.write(f'# total kernel time in window {tot/1e3:.1f} ms over '
                f'{kept_span:.2f} s wall\n')
        f.write(f"{'pct':>7} {'calls':>7} {'avg_us':>10}  kernel\n")
        for name, (n, us) in sorted(agg.items(), key=lambda kv: -kv[1][1])[:30]:
            f.write(f'{us/tot*100:6.2f}% {n:7d} {us/n:10.2f}  {name[:110]}\n')
    print(open(out).read().split('kernel\n')[0])

clean_table('inf', R + '/gpurun_out/infer_kernels_clean.txt', 80, 60)
clean_table('tr', R + '/gpurun_out/train_kernels_clean.txt', 17, 12)
PYEOF
head -14 $R/gpurun_out/infer_kernels_clean.txt
head -14 $R/gpurun_out/train_kernels_clean.txt
