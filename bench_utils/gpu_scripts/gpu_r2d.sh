#!/bin/bash
# Round-2 call D: full gpu pytest (fused attender, NLL-lse, 80-step graph
# equivalence), attnlnp2d TunableOp bisect, RCCL smoke, fused-attender bench,
# cb2d-change bench, eager-vs-graph head-to-head calibration.
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
export PYTHONUNBUFFERED=1

timeout 900 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu_r2d.log 2>&1
echo "pytest-gpu: $?" | tee gpurun_out/summary_r2d.txt
tail -3 gpurun_out/pytest_gpu_r2d.log >> gpurun_out/summary_r2d.txt

NPF_BENCH_NO_TUNE=1 timeout 300 python bench.py --model attnlnp2d --steps 20 --warmup 5 --no-graph \
  > gpurun_out/lnp_notune.json 2> gpurun_out/lnp_notune.log
echo "lnp-notune-nograph: $?" | tee -a gpurun_out/summary_r2d.txt
cat gpurun_out/lnp_notune.json >> gpurun_out/summary_r2d.txt

NPF_BENCH_NO_TUNE=1 timeout 300 python bench.py --model attnlnp2d --steps 50 --warmup 10 \
  > gpurun_out/lnp_notune_g.json 2> gpurun_out/lnp_notune_g.log
echo "lnp-notune-graph: $?" | tee -a gpurun_out/summary_r2d.txt
cat gpurun_out/lnp_notune_g.json >> gpurun_out/summary_r2d.txt
grep -i graph gpurun_out/lnp_notune_g.log >> gpurun_out/summary_r2d.txt

timeout 300 python bench_utils/rccl_smoke.py all > gpurun_out/rccl_smoke.log 2>&1
echo "rccl-smoke: $?" | tee -a gpurun_out/summary_r2d.txt
tail -3 gpurun_out/rccl_smoke.log >> gpurun_out/summary_r2d.txt

timeout 420 python bench.py --model attncnp --steps 300 --warmup 30 \
  > gpurun_out/bench_attncnp_fused.json 2> gpurun_out/bench_attncnp_fused.log
echo "bench-attncnp-fusedattender: $?" | tee -a gpurun_out/summary_r2d.txt
cat gpurun_out/bench_attncnp_fused.json >> gpurun_out/summary_r2d.txt

timeout 420 python bench.py --model gridconvlnp2d --steps 40 --warmup 8 \
  > gpurun_out/bench_glnp_dw.json 2> gpurun_out/bench_glnp_dw.log
echo "bench-gridconvlnp-dwloop: $?" | tee -a gpurun_out/summary_r2d.txt
cat gpurun_out/bench_glnp_dw.json >> gpurun_out/summary_r2d.txt

# head-to-head: identical budget, eager vs graphed
timeout 420 python examples/train_gp_1d.py --models AttnCNP --datasets RBF_Kernel \
  --epochs 1 --n-tasks 50000 --device-episodes --bf16 \
  --chckpnt-dir gpurun_out/h2h_eager/ > gpurun_out/h2h_eager.log 2>&1
echo "h2h-eager: $?" | tee -a gpurun_out/summary_r2d.txt
grep -E "epoch|test log" gpurun_out/h2h_eager.log | tail -2 >> gpurun_out/summary_r2d.txt

timeout 420 python examples/train_gp_1d.py --models AttnCNP --datasets RBF_Kernel \
  --epochs 1 --n-tasks 50000 --device-episodes --bf16 --hipgraphs \
  --chckpnt-dir gpurun_out/h2h_graph/ > gpurun_out/h2h_graph.log 2>&1
echo "h2h-graph: $?" | tee -a gpurun_out/summary_r2d.txt
grep -E "epoch|test log" gpurun_out/h2h_graph.log | tail -2 >> gpurun_out/summary_r2d.txt
