#!/bin/bash
# Round-2 call C: attnlnp2d fault bisect + graph-equivalence test + RCCL v2
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
export PYTHONUNBUFFERED=1

timeout 240 python bench_utils/lnp_crash_diag.py eagerops > gpurun_out/diag_eagerops.log 2>&1
echo "diag-eagerops: $?" | tee gpurun_out/summary_r2c.txt
tail -3 gpurun_out/diag_eagerops.log >> gpurun_out/summary_r2c.txt

timeout 240 python bench_utils/lnp_crash_diag.py fused > gpurun_out/diag_fused.log 2>&1
echo "diag-fused: $?" | tee -a gpurun_out/summary_r2c.txt
tail -3 gpurun_out/diag_fused.log >> gpurun_out/summary_r2c.txt

timeout 420 python -m pytest tests/test_trainer.py -m gpu -q -x > gpurun_out/graph_equiv.log 2>&1
echo "graph-equiv: $?" | tee -a gpurun_out/summary_r2c.txt
tail -5 gpurun_out/graph_equiv.log >> gpurun_out/summary_r2c.txt

timeout 300 python bench_utils/rccl_smoke.py all > gpurun_out/rccl_smoke.log 2>&1
echo "rccl-smoke: $?" | tee -a gpurun_out/summary_r2c.txt
tail -4 gpurun_out/rccl_smoke.log >> gpurun_out/summary_r2c.txt

# trainer divergence bisect: fused Adam + device episodes, NO graphs
timeout 600 python examples/train_gp_1d.py --models AttnCNP --datasets RBF_Kernel \
  --epochs 2 --n-tasks 20000 --device-episodes --bf16 \
  --chckpnt-dir gpurun_out/calib3/ > gpurun_out/calib3.log 2>&1
echo "calib-nograph-fusedadam: $?" | tee -a gpurun_out/summary_r2c.txt
grep -E "epoch|test log" gpurun_out/calib3.log | tail -4 >> gpurun_out/summary_r2c.txt
