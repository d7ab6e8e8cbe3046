#!/bin/bash
# Round-2 batch E: block fusion parity + whole-step effect; 8ph MINK=256
# full table; pooled pipe evidence.
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

echo "== block fusion parity ==" | tee gpurun_out/r2e.log
timeout 600 python -m pytest tests/test_block_fusion_gpu.py tests/test_convergence_gpu.py -q 2>&1 | tail -3 | tee -a gpurun_out/r2e.log

echo "== r50 bench b2048 (block fusion) ==" | tee -a gpurun_out/r2e.log
timeout 500 python bench.py --gpus 1 --steps 20 --warmup 8 2>&1 | tail -1 | tee -a gpurun_out/r2e.log
echo "== r50 bench b256 (block fusion) ==" | tee -a gpurun_out/r2e.log
timeout 500 python bench.py --gpus 1 --steps 20 --warmup 8 --batch-size 256 2>&1 | tail -1 | tee -a gpurun_out/r2e.log

echo "== r50 bench b2048 MINK=256 ==" | tee -a gpurun_out/r2e.log
DLX_8PH_MINK=256 timeout 500 python bench.py --gpus 1 --steps 20 --warmup 8 2>&1 | tail -1 | tee -a gpurun_out/r2e.log

echo "== conv shapes b1024 MINK=256 ==" | tee -a gpurun_out/r2e.log
DLX_8PH_MINK=256 timeout 600 python tools/bench_conv_shapes.py --batch 1024 > gpurun_out/conv_shapes_b1024_mink256.txt 2>&1
tail -5 gpurun_out/conv_shapes_b1024_mink256.txt | tee -a gpurun_out/r2e.log

echo "== pooled pipe throughput ==" | tee -a gpurun_out/r2e.log
timeout 300 python tools/bench_pipe.py 2>&1 | tee -a gpurun_out/r2e.log
