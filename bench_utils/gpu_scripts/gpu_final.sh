#!/bin/bash
set -x
mkdir -p gpurun_out
R=$GRAFT_REPO_ROOT
rm -rf gpurun_out/final
mkdir -p gpurun_out/final
timeout 120 python __graft_entry__.py smoke > gpurun_out/final/smoke.log 2>&1
echo "smoke: $?" | tee gpurun_out/final/summary.txt
timeout 600 python -m pytest tests -m gpu -q > gpurun_out/final/pytest.log 2>&1
echo "pytest-gpu: $?" | tee -a gpurun_out/final/summary.txt
tail -2 gpurun_out/final/pytest.log >> gpurun_out/final/summary.txt
for m in attncnp convcnp attnlnp2d gridconvlnp2d; do
  case $m in attncnp|convcnp) steps=300;; attnlnp2d) steps=120;; *) steps=40;; esac
  timeout 300 python bench.py --model $m --steps $steps --warmup 20 > gpurun_out/final/bench_$m.json 2>/dev/null
  echo "bench-$m: $?" | tee -a gpurun_out/final/summary.txt
  cat gpurun_out/final/bench_$m.json >> gpurun_out/final/summary.txt
done
cd /tmp && export TMPDIR=/tmp NPF_BENCH_NO_TUNE=1
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d $R/gpurun_out/final -o attncnp_final -- python $R/bench.py --steps 60 --warmup 15 > /dev/null 2>&1
echo "prof-attncnp: $?" >> $R/gpurun_out/final/summary.txt
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d $R/gpurun_out/final -o attnlnp2d_final -- python $R/bench.py --model attnlnp2d --steps 50 --warmup 10 > /dev/null 2>&1
echo "prof-attnlnp2d: $?" >> $R/gpurun_out/final/summary.txt
cd $R
find gpurun_out/final -name '*kernel_trace.csv' -delete
cat gpurun_out/final/summary.txt
