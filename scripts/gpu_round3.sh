set -x
mkdir -p gpurun_out
R=$GRAFT_REPO_ROOT
timeout 420 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest_gpu_exit=$?" >> gpurun_out/pytest_gpu.log
timeout 300 python bench.py --mode train --steps 15 --warmup 5 > gpurun_out/bench_train.log 2>&1
timeout 300 python bench.py --mode infer --steps 30 --warmup 10 > gpurun_out/bench_infer.log 2>&1
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats -d $R/gpurun_out/prof -o train2 -- python $R/bench.py --mode train --steps 5 --warmup 2 > $R/gpurun_out/rocprof_train.log 2>&1
tail -2 $R/gpurun_out/bench_train.log; tail -2 $R/gpurun_out/bench_infer.log; tail -3 $R/gpurun_out/pytest_gpu.log
