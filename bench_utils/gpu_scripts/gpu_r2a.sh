#!/bin/bash
# Round-2 call A: validate round-2 changes on hardware.
# 1) pytest -m gpu  2) RCCL 2-ranks-on-1-GPU smoke  3) latent graph capture
# 4) device-episodes training calibration (2 epochs AttnCNP)
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

timeout 600 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu_r2a.log 2>&1
echo "pytest-gpu: $?" | tee gpurun_out/summary_r2a.txt
tail -3 gpurun_out/pytest_gpu_r2a.log >> gpurun_out/summary_r2a.txt

# RCCL on hardware: two ranks sharing cuda:0 — proves init, flat
# all-reduce, graph-fallback ladder and replica sync on the real backend
timeout 420 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29571 \
  bench.py --gpus 2 --steps 20 --warmup 5 > gpurun_out/rccl2_bench.json 2> gpurun_out/rccl2.log
echo "rccl2: $?" | tee -a gpurun_out/summary_r2a.txt
cat gpurun_out/rccl2_bench.json >> gpurun_out/summary_r2a.txt
tail -5 gpurun_out/rccl2.log >> gpurun_out/summary_r2a.txt

# latent models: noise pool => hipGraph capture should now succeed
timeout 420 python bench.py --model attnlnp2d --steps 100 --warmup 15 \
  > gpurun_out/bench_attnlnp2d_r2.json 2> gpurun_out/bench_attnlnp2d_r2.log
echo "attnlnp2d: $?" | tee -a gpurun_out/summary_r2a.txt
cat gpurun_out/bench_attnlnp2d_r2.json >> gpurun_out/summary_r2a.txt
grep -i graph gpurun_out/bench_attnlnp2d_r2.log >> gpurun_out/summary_r2a.txt

timeout 420 python bench.py --model gridconvlnp2d --steps 40 --warmup 8 \
  > gpurun_out/bench_gridconvlnp2d_r2.json 2> gpurun_out/bench_gridconvlnp2d_r2.log
echo "gridconvlnp2d: $?" | tee -a gpurun_out/summary_r2a.txt
cat gpurun_out/bench_gridconvlnp2d_r2.json >> gpurun_out/summary_r2a.txt
grep -i graph gpurun_out/bench_gridconvlnp2d_r2.log >> gpurun_out/summary_r2a.txt

# training calibration: 2 epochs of the reference AttnCNP budget shape
timeout 600 python examples/train_gp_1d.py --models AttnCNP --datasets RBF_Kernel \
  --epochs 2 --n-tasks 50000 --device-episodes --bf16 \
  --chckpnt-dir gpurun_out/calib/ > gpurun_out/calib_attncnp.log 2>&1
echo "calib: $?" | tee -a gpurun_out/summary_r2a.txt
tail -6 gpurun_out/calib_attncnp.log >> gpurun_out/summary_r2a.txt
