#!/bin/bash
set -x
mkdir -p gpurun_out
R=$GRAFT_REPO_ROOT
rm -rf gpurun_out/prof
timeout 600 python bench.py --steps 600 --warmup 50 > gpurun_out/bench.json 2> gpurun_out/bench.log
echo "bench600: $?" | tee gpurun_out/summary.txt
cat gpurun_out/bench.json | tee -a gpurun_out/summary.txt
timeout 300 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest-gpu: $?" | tee -a gpurun_out/summary.txt
tail -2 gpurun_out/pytest_gpu.log >> gpurun_out/summary.txt
cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats -d $R/gpurun_out/prof -o attncnp -- python $R/bench.py --steps 100 --warmup 30 > $R/gpurun_out/bench_prof.json 2> $R/gpurun_out/prof.log
echo "rocprof: $?" >> $R/gpurun_out/summary.txt
cd $R
# keep only the small stats/summary files: the merge-back cap is 64 MiB
find gpurun_out/prof -type f -size +4M -delete
find gpurun_out/prof -name '*kernel_trace*' -delete
du -sh gpurun_out >> gpurun_out/summary.txt
cat gpurun_out/summary.txt
