#!/bin/bash
# Round: all-model GPU validation + demangled step profile.
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
export DLX_EFS=${DLX_EFS:-/tmp/dlx_efs}

echo "== pytest gpu ==" | tee gpurun_out/p7.log
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -3 | tee -a gpurun_out/p7.log

echo "== bench r50 ==" | tee -a gpurun_out/p7.log
timeout 500 python bench.py --gpus 1 --steps 20 --warmup 8 2>&1 | tail -1 | tee -a gpurun_out/p7.log

echo "== bert-base train 12 steps ==" | tee -a gpurun_out/p7.log
timeout 500 python -m deeplearning_cfn_amd.train --network bert \
  --batch-size 32 --seq-len 128 --steps-per-epoch 12 --disp-batches 4 \
  --num-epochs 1 --checkpoint-every 0 --lr 1e-4 2>&1 | tail -5 | tee -a gpurun_out/p7.log

echo "== resnet152 train 8 steps (288GB sizing) ==" | tee -a gpurun_out/p7.log
timeout 500 python -m deeplearning_cfn_amd.train --network resnet \
  --num-layers 152 --batch-size 256 --steps-per-epoch 8 --disp-batches 4 \
  --num-epochs 1 --checkpoint-every 0 2>&1 | tail -4 | tee -a gpurun_out/p7.log

echo "== cifar resnet20 train 20 steps ==" | tee -a gpurun_out/p7.log
timeout 300 python -m deeplearning_cfn_amd.train --network resnet \
  --num-layers 20 --dataset cifar10 --batch-size 512 --steps-per-epoch 20 \
  --disp-batches 10 --num-epochs 1 --checkpoint-every 0 2>&1 | tail -4 | tee -a gpurun_out/p7.log

echo "== stats profile (demangled) ==" | tee -a gpurun_out/p7.log
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
timeout 500 rocprofv3 --kernel-trace --stats -d /tmp/prof_p7 -o p7 -- \
  python bench.py --gpus 1 --steps 5 --warmup 5 2>&1 | tail -1 | tee -a gpurun_out/p7.log
python tools/prof_summary.py /tmp/prof_p7/p7_results.db \
  gpurun_out/r50_step_kernels_p7.md "ResNet-50 b256 bf16 native (wrw-128sq), 5+5 steps, full names" \
  | tee -a gpurun_out/p7.log
du -sh gpurun_out | tee -a gpurun_out/p7.log
