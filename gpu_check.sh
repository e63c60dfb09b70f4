#!/bin/bash
# round-1 GPU validation: smoke + gpu tests + bench + rocprof stats
set -x
mkdir -p gpurun_out
R=$GRAFT_REPO_ROOT
python __graft_entry__.py smoke > gpurun_out/smoke.log 2>&1
echo "smoke: $?" | tee gpurun_out/summary.txt
timeout 900 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest-gpu: $?" | tee -a gpurun_out/summary.txt
tail -3 gpurun_out/pytest_gpu.log >> gpurun_out/summary.txt
timeout 600 python bench.py --steps 300 --warmup 50 > gpurun_out/bench.json 2> gpurun_out/bench.log
echo "bench: $?" | tee -a gpurun_out/summary.txt
cat gpurun_out/bench.json >> gpurun_out/summary.txt
cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats -d $R/gpurun_out/prof -o attncnp -- python $R/bench.py --steps 100 --warmup 30 > $R/gpurun_out/bench_prof.json 2> $R/gpurun_out/prof.log
echo "rocprof: $?" >> $R/gpurun_out/summary.txt
cat $R/gpurun_out/summary.txt
