#!/bin/bash
# Round-2 call H: profiles (stats only, traces deleted), small artifacts
# rerun (lost to the 64MiB cap), cb2d microbench, final flagship bench.
set -x
mkdir -p gpurun_out/r2
R=$GRAFT_REPO_ROOT
export HSA_ENABLE_IPC_MODE_LEGACY=0 PYTHONUNBUFFERED=1

timeout 240 python bench_utils/cb2d_micro.py > gpurun_out/r2/cb2d_micro.log 2>&1
echo "cb2d-micro: $?" | tee gpurun_out/r2/summary.txt

timeout 420 python bench.py --model attncnp --steps 300 --warmup 30 \
  > gpurun_out/r2/bench_attncnp.json 2>/dev/null
echo "bench-attncnp: $?" | tee -a gpurun_out/r2/summary.txt
cat gpurun_out/r2/bench_attncnp.json >> gpurun_out/r2/summary.txt

cd /tmp && export TMPDIR=/tmp
cd $R
NPF_BENCH_NO_TUNE=1 timeout 360 rocprofv3 --kernel-trace --stats --output-format csv \
  -d $R/gpurun_out/r2 -o attncnp_r2 -- python bench.py --model attncnp --steps 60 --warmup 10 \
  > /dev/null 2> gpurun_out/r2/prof_att.log
echo "prof-attncnp: $?" | tee -a gpurun_out/r2/summary.txt
NPF_BENCH_NO_TUNE=1 timeout 360 rocprofv3 --kernel-trace --stats --output-format csv \
  -d $R/gpurun_out/r2 -o gridconvlnp_r2 -- python bench.py --model gridconvlnp2d --steps 15 --warmup 4 \
  > /dev/null 2> gpurun_out/r2/prof_glnp.log
echo "prof-glnp: $?" | tee -a gpurun_out/r2/summary.txt
find gpurun_out/r2 -name '*kernel_trace.csv' -delete

timeout 300 python -m npf.cli train --model LNP --data RBF_Kernel \
  --epochs 3 --n-tasks 2000 --n-test-tasks 500 --batch-size 32 --bf16 \
  --device-episodes --loss elbo --min-sigma-pred 0.1 --min-lat 0.1 \
  --chckpnt-dir gpurun_out/r2/ablation/minsig0.1_minlat0.1/ \
  --data-cache /tmp/abl_cache.npz > gpurun_out/r2/abl1.log 2>&1
echo "ablation-1: $?" | tee -a gpurun_out/r2/summary.txt
timeout 300 python -m npf.cli train --model LNP --data RBF_Kernel \
  --epochs 3 --n-tasks 2000 --n-test-tasks 500 --batch-size 32 --bf16 \
  --device-episodes --loss nll \
  --chckpnt-dir gpurun_out/r2/ablation/npml_dflt/ \
  --data-cache /tmp/abl_cache.npz > gpurun_out/r2/abl2.log 2>&1
echo "ablation-2: $?" | tee -a gpurun_out/r2/summary.txt
grep -hE "test log" gpurun_out/r2/abl1.log gpurun_out/r2/abl2.log >> gpurun_out/r2/summary.txt

timeout 300 python examples/train_img_2d.py --models GridConvCNP --datasets synthetic32 \
  --epochs 5 --bf16 --device-episodes \
  --chckpnt-dir gpurun_out/r2/img_partial/ > gpurun_out/r2/img_partial.log 2>&1
echo "img-partial: $?" | tee -a gpurun_out/r2/summary.txt
grep -E "test log" gpurun_out/r2/img_partial.log >> gpurun_out/r2/summary.txt
tail -6 gpurun_out/r2/cb2d_micro.log >> gpurun_out/r2/summary.txt
du -sh gpurun_out
