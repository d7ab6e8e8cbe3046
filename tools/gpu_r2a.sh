#!/bin/bash
# Round-2 batch A: attention + LN kernels on HW, BERT A/B, r50 regression.
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

echo "== attention + kernels2 (LN) ==" | tee gpurun_out/r2a.log
timeout 600 python -m pytest tests/test_attention_gpu.py \
  tests/test_kernels2_gpu.py tests/test_kernels_gpu.py -q 2>&1 | tail -5 | tee -a gpurun_out/r2a.log

echo "== bert-base b32 s128 fused attn (default) ==" | tee -a gpurun_out/r2a.log
timeout 400 python -m deeplearning_cfn_amd.train --network bert \
  --batch-size 32 --seq-len 128 --steps-per-epoch 12 --disp-batches 4 \
  --num-epochs 1 --checkpoint-every 0 --eval-steps 0 --lr 1e-4 2>&1 | tail -4 | tee -a gpurun_out/r2a.log

echo "== bert-base b32 s128 bmm attn (DLX_ATTN=0) ==" | tee -a gpurun_out/r2a.log
DLX_ATTN=0 timeout 400 python -m deeplearning_cfn_amd.train --network bert \
  --batch-size 32 --seq-len 128 --steps-per-epoch 12 --disp-batches 4 \
  --num-epochs 1 --checkpoint-every 0 --eval-steps 0 --lr 1e-4 2>&1 | tail -4 | tee -a gpurun_out/r2a.log

echo "== bert-base b16 s512 fused ==" | tee -a gpurun_out/r2a.log
timeout 400 python -m deeplearning_cfn_amd.train --network bert \
  --batch-size 16 --seq-len 512 --steps-per-epoch 8 --disp-batches 4 \
  --num-epochs 1 --checkpoint-every 0 --eval-steps 0 --lr 1e-4 2>&1 | tail -4 | tee -a gpurun_out/r2a.log

echo "== bert-base b16 s512 bmm ==" | tee -a gpurun_out/r2a.log
DLX_ATTN=0 timeout 400 python -m deeplearning_cfn_amd.train --network bert \
  --batch-size 16 --seq-len 512 --steps-per-epoch 8 --disp-batches 4 \
  --num-epochs 1 --checkpoint-every 0 --eval-steps 0 --lr 1e-4 2>&1 | tail -4 | tee -a gpurun_out/r2a.log

echo "== r50 bench regression check ==" | tee -a gpurun_out/r2a.log
timeout 500 python bench.py --gpus 1 --steps 20 --warmup 8 2>&1 | tail -1 | tee -a gpurun_out/r2a.log
