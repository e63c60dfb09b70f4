#!/bin/bash
mkdir -p gpurun_out
exec > gpurun_out/fix1.log 2>&1
set -x
timeout 600 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest-gpu: $?"
tail -4 gpurun_out/pytest_gpu.log
timeout 300 python bench.py --model convcnp --steps 200 --warmup 30 > gpurun_out/bench_convcnp.json 2>&1
echo "bench-convcnp: $?"; cat gpurun_out/bench_convcnp.json | tail -1
timeout 300 python bench.py --model gridconvlnp2d --steps 60 --warmup 15 > gpurun_out/bench_gridconvlnp2d.json 2>&1
echo "bench-gridconvlnp2d: $?"; cat gpurun_out/bench_gridconvlnp2d.json | tail -1
timeout 420 python bench_utils/lnp_trace.py all
echo "lnp-trace: $?"
