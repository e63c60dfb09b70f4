#!/bin/bash
# Round-2 call B: attnlnp2d fault bisect, RCCL retry, graphed-trainer calib.
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
export PYTHONUNBUFFERED=1

# 1) bisect: graph+pool (crashed in call A) with serialized kernels
AMD_SERIALIZE_KERNEL=3 timeout 300 python bench.py --model attnlnp2d --steps 5 --warmup 3 \
  > gpurun_out/lnp_g.json 2> gpurun_out/lnp_g.log
echo "lnp-graphpool: $?" | tee gpurun_out/summary_r2b.txt
tail -4 gpurun_out/lnp_g.log >> gpurun_out/summary_r2b.txt

# 2) no-graph (round-1 behavior): should pass
timeout 300 python bench.py --model attnlnp2d --steps 5 --warmup 3 --no-graph \
  > gpurun_out/lnp_e.json 2> gpurun_out/lnp_e.log
echo "lnp-eager: $?" | tee -a gpurun_out/summary_r2b.txt

# 3) RCCL smoke retry (init_distributed modulo fix)
timeout 420 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
  --master-addr 127.0.0.1 --master-port 29573 \
  bench.py --gpus 2 --steps 20 --warmup 5 > gpurun_out/rccl2_bench.json 2> gpurun_out/rccl2.log
echo "rccl2: $?" | tee -a gpurun_out/summary_r2b.txt
cat gpurun_out/rccl2_bench.json >> gpurun_out/summary_r2b.txt

# 4) graphed trainer calibration (2 epochs, reference budget shape)
timeout 800 python examples/train_gp_1d.py --models AttnCNP --datasets RBF_Kernel \
  --epochs 2 --n-tasks 50000 --device-episodes --bf16 --hipgraphs \
  --chckpnt-dir gpurun_out/calib2/ > gpurun_out/calib2.log 2>&1
echo "calib2: $?" | tee -a gpurun_out/summary_r2b.txt
grep -E "epoch|loss|Error|error" gpurun_out/calib2.log | tail -8 >> gpurun_out/summary_r2b.txt
