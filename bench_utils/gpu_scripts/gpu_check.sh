#!/bin/bash
set -x
mkdir -p gpurun_out
R=$GRAFT_REPO_ROOT
rm -rf gpurun_out/prof
timeout 600 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest-gpu: $?" | tee -a gpurun_out/summary.txt
tail -2 gpurun_out/pytest_gpu.log >> gpurun_out/summary.txt
for m in attncnp convcnp attnlnp2d gridconvlnp2d; do
  case $m in
    attncnp|convcnp) steps=300;;
    attnlnp2d) steps=150;;
    gridconvlnp2d) steps=50;;
  esac
  timeout 480 python bench.py --model $m --steps $steps --warmup 20 > gpurun_out/bench_$m.json 2> gpurun_out/bench_$m.log
  echo "bench-$m: $?" | tee -a gpurun_out/summary.txt
  cat gpurun_out/bench_$m.json >> gpurun_out/summary.txt
done
cd /tmp && export TMPDIR=/tmp
export NPF_BENCH_NO_TUNE=1
for m in attncnp convcnp attnlnp2d; do
  timeout 420 rocprofv3 --kernel-trace --stats --output-format csv -d $R/gpurun_out/prof -o $m -- python $R/bench.py --model $m --steps 60 --warmup 15 > $R/gpurun_out/prof_$m.json 2> $R/gpurun_out/prof_$m.log
  echo "rocprof-$m: $?" >> $R/gpurun_out/summary.txt
done
cd $R
find gpurun_out/prof -name '*kernel_trace.csv' -delete
find gpurun_out/prof -type f -size +8M -delete
cat gpurun_out/summary.txt
