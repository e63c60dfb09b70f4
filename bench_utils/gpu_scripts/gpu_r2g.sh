#!/bin/bash
# Round-2 call G: validation + perf matrix + profiles + ablation smoke.
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
export PYTHONUNBUFFERED=1

timeout 900 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu_r2g.log 2>&1
echo "pytest-gpu: $?" | tee gpurun_out/summary_r2g.txt
tail -2 gpurun_out/pytest_gpu_r2g.log >> gpurun_out/summary_r2g.txt

# flagship A/B: tuned + fused attender vs tuned + composed
timeout 420 python bench.py --model attncnp --steps 300 --warmup 30 \
  > gpurun_out/b_att_fused.json 2>/dev/null
echo "attncnp-fused: $? $(cat gpurun_out/b_att_fused.json | python3 -c 'import json,sys; d=json.load(sys.stdin); print(round(d["value"]),d["ms_per_step"])')" | tee -a gpurun_out/summary_r2g.txt
NPF_NO_FUSED_ATTENDER=1 timeout 420 python bench.py --model attncnp --steps 300 --warmup 30 \
  > gpurun_out/b_att_nofused.json 2>/dev/null
echo "attncnp-nofused: $? $(cat gpurun_out/b_att_nofused.json | python3 -c 'import json,sys; d=json.load(sys.stdin); print(round(d["value"]),d["ms_per_step"])')" | tee -a gpurun_out/summary_r2g.txt

timeout 420 python bench.py --model convcnp --steps 300 --warmup 30 \
  > gpurun_out/b_convcnp_r2.json 2>/dev/null
echo "convcnp: $? $(cat gpurun_out/b_convcnp_r2.json | python3 -c 'import json,sys; d=json.load(sys.stdin); print(round(d["value"]),d["ms_per_step"])')" | tee -a gpurun_out/summary_r2g.txt

timeout 420 python bench.py --model gridconvlnp2d --steps 60 --warmup 10 \
  > gpurun_out/b_glnp_r2.json 2>/dev/null
echo "gridconvlnp2d: $? $(cat gpurun_out/b_glnp_r2.json | python3 -c 'import json,sys; d=json.load(sys.stdin); print(round(d["value"]),d["ms_per_step"])')" | tee -a gpurun_out/summary_r2g.txt

timeout 420 python bench.py --model attnlnp2d --steps 100 --warmup 15 \
  > gpurun_out/b_alnp_r2.json 2>/dev/null
echo "attnlnp2d: $? $(cat gpurun_out/b_alnp_r2.json | python3 -c 'import json,sys; d=json.load(sys.stdin); print(round(d["value"]),d["ms_per_step"])')" | tee -a gpurun_out/summary_r2g.txt

# rocprof kernel stats for the two headline perf configs
cd /tmp && export TMPDIR=/tmp && cd $GRAFT_REPO_ROOT
NPF_BENCH_NO_TUNE=1 timeout 420 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_att -- \
  python bench.py --model attncnp --steps 60 --warmup 10 > gpurun_out/prof_att.json 2> gpurun_out/prof_att.log
echo "prof-attncnp: $?" | tee -a gpurun_out/summary_r2g.txt
NPF_BENCH_NO_TUNE=1 timeout 420 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_glnp -- \
  python bench.py --model gridconvlnp2d --steps 20 --warmup 5 > gpurun_out/prof_glnp.json 2> gpurun_out/prof_glnp.log
echo "prof-glnp: $?" | tee -a gpurun_out/summary_r2g.txt

# loss-ablation smoke: two grid points of the Losses.ipynb grid
timeout 420 python -m npf.cli train --model LNP --data RBF_Kernel \
  --epochs 3 --n-tasks 2000 --n-test-tasks 500 --batch-size 32 --bf16 \
  --device-episodes --loss elbo --min-sigma-pred 0.1 --min-lat 0.1 \
  --chckpnt-dir gpurun_out/ablation/minsig0.1_minlat0.1/ \
  --data-cache gpurun_out/abl_cache.npz > gpurun_out/abl1.log 2>&1
echo "ablation-1: $?" | tee -a gpurun_out/summary_r2g.txt
timeout 420 python -m npf.cli train --model LNP --data RBF_Kernel \
  --epochs 3 --n-tasks 2000 --n-test-tasks 500 --batch-size 32 --bf16 \
  --device-episodes --loss nll \
  --chckpnt-dir gpurun_out/ablation/npml_dflt/ \
  --data-cache gpurun_out/abl_cache.npz > gpurun_out/abl2.log 2>&1
echo "ablation-2: $?" | tee -a gpurun_out/summary_r2g.txt
grep -E "test log" gpurun_out/abl1.log gpurun_out/abl2.log >> gpurun_out/summary_r2g.txt

# partial-budget GridConvCNP on synthetic images (documented: no real data)
timeout 420 python examples/train_img_2d.py --models GridConvCNP --datasets synthetic32 \
  --epochs 5 --bf16 --device-episodes \
  --chckpnt-dir gpurun_out/img_partial/ > gpurun_out/img_partial.log 2>&1
echo "img-partial: $?" | tee -a gpurun_out/summary_r2g.txt
grep -E "epoch|test log" gpurun_out/img_partial.log | tail -3 >> gpurun_out/summary_r2g.txt
