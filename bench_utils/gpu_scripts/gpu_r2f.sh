#!/bin/bash
# Round-2 call F: headline training in fp32 (bf16 destabilizes the
# high-precision LL regime; the reference trains fp32).
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
export PYTHONUNBUFFERED=1

timeout 900 python examples/train_gp_1d.py --models AttnCNP --datasets RBF_Kernel \
  --epochs 100 --n-tasks 50000 --device-episodes --hipgraphs \
  --chckpnt-dir gpurun_out/trained_fp32/ > gpurun_out/train_attncnp_fp32.log 2>&1
echo "train-attncnp-fp32: $?" | tee gpurun_out/summary_r2f.txt
grep -E "epoch (1|10|25|50|75|100)/100|test log" gpurun_out/train_attncnp_fp32.log | tail -8 >> gpurun_out/summary_r2f.txt

timeout 1100 python examples/train_gp_1d.py --models ConvCNP --datasets Periodic_Kernel \
  --epochs 100 --n-tasks 50000 --device-episodes --hipgraphs \
  --chckpnt-dir gpurun_out/trained_fp32/ > gpurun_out/train_convcnp_fp32.log 2>&1
echo "train-convcnp-fp32: $?" | tee -a gpurun_out/summary_r2f.txt
grep -E "epoch (1|10|25|50|75|100)/100|test log" gpurun_out/train_convcnp_fp32.log | tail -8 >> gpurun_out/summary_r2f.txt
