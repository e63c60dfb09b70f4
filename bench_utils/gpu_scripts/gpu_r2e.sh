#!/bin/bash
# Round-2 call E: THE headline runs — train to the reference budget on
# MI355X and evaluate on the fixed 10k-task test chunks.
#   AttnCNP on RBF:      100 epochs x 50k fresh tasks, batch 32, Adam 1e-3,
#                        x10 decay, seed 123  (published LL: 149.16)
#   ConvCNP on Periodic: same budget            (published LL: 192.97)
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
export PYTHONUNBUFFERED=1

timeout 900 python examples/train_gp_1d.py --models AttnCNP --datasets RBF_Kernel \
  --epochs 100 --n-tasks 50000 --device-episodes --bf16 --hipgraphs \
  --chckpnt-dir gpurun_out/trained/ > gpurun_out/train_attncnp_rbf.log 2>&1
echo "train-attncnp-rbf: $?" | tee gpurun_out/summary_r2e.txt
grep -E "epoch (1|25|50|75|100)/100|test log" gpurun_out/train_attncnp_rbf.log | tail -8 >> gpurun_out/summary_r2e.txt

timeout 1100 python examples/train_gp_1d.py --models ConvCNP --datasets Periodic_Kernel \
  --epochs 100 --n-tasks 50000 --device-episodes --bf16 --hipgraphs \
  --chckpnt-dir gpurun_out/trained/ > gpurun_out/train_convcnp_periodic.log 2>&1
echo "train-convcnp-periodic: $?" | tee -a gpurun_out/summary_r2e.txt
grep -E "epoch (1|25|50|75|100)/100|test log" gpurun_out/train_convcnp_periodic.log | tail -8 >> gpurun_out/summary_r2e.txt
