#!/bin/bash
# Round-2 call J: train CNP and LNP to the reference budget (fp32, graphed).
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0 PYTHONUNBUFFERED=1

timeout 700 python examples/train_gp_1d.py --models CNP --datasets RBF_Kernel \
  --epochs 100 --n-tasks 50000 --device-episodes --hipgraphs \
  --chckpnt-dir gpurun_out/trained_j/ > gpurun_out/train_cnp.log 2>&1
echo "train-cnp: $?" | tee gpurun_out/summary_j.txt
grep -E "epoch (1|25|50|100)/100|test log" gpurun_out/train_cnp.log | tail -5 >> gpurun_out/summary_j.txt

timeout 700 python examples/train_gp_1d.py --models LNP --datasets RBF_Kernel \
  --epochs 100 --n-tasks 50000 --device-episodes --hipgraphs \
  --chckpnt-dir gpurun_out/trained_j/ > gpurun_out/train_lnp.log 2>&1
echo "train-lnp: $?" | tee -a gpurun_out/summary_j.txt
grep -E "epoch (1|25|50|100)/100|test log" gpurun_out/train_lnp.log | tail -5 >> gpurun_out/summary_j.txt
