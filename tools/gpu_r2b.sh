#!/bin/bash
# Round-2 batch B: native stem conv parity + perf, attention bwd A/B redo.
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

echo "== conv parity (incl stem shapes) ==" | tee gpurun_out/r2b.log
timeout 600 python -m pytest tests/test_conv_gpu.py -q 2>&1 | tail -4 | tee -a gpurun_out/r2b.log

echo "== stem timing vs MIOpen (b256) ==" | tee -a gpurun_out/r2b.log
timeout 600 python - <<'EOF' 2>&1 | tee -a gpurun_out/r2b.log
import torch, time, sys
sys.path.insert(0, ".")
from deeplearning_cfn_amd.ops import native
n = native()
def t(fn, iters=10, warm=3):
    for _ in range(warm): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/iters*1e3
b = 256
x = torch.randn(b,3,224,224,device="cuda").bfloat16().contiguous(memory_format=torch.channels_last)
w = (torch.randn(64,3,7,7,device="cuda")*0.05).bfloat16()
dy = torch.randn(b,64,112,112,device="cuda").bfloat16().contiguous(memory_format=torch.channels_last)
print(f"stem fwd native  : {t(lambda: n.conv2d_fwd(x,w,2,3)):.3f} ms")
print(f"stem fwd miopen  : {t(lambda: torch.nn.functional.conv2d(x,w.contiguous(memory_format=torch.channels_last),None,2,3)):.3f} ms")
print(f"stem wrw native  : {t(lambda: n.conv2d_wrw(dy,x,7,7,2,3)):.3f} ms")
print(f"stem wrw miopen  : {t(lambda: torch.ops.aten.convolution_backward(dy,x,w,None,[2,2],[3,3],[1,1],False,[0,0],1,[False,True,False])[1]):.3f} ms")
print(f"stem fwd_stats   : {t(lambda: n.conv2d_fwd_stats(x,w,2,3)):.3f} ms")
EOF

echo "== bert b32 s128 fused attn (bf16 bwd) ==" | tee -a gpurun_out/r2b.log
timeout 400 python -m deeplearning_cfn_amd.train --network bert \
  --batch-size 32 --seq-len 128 --steps-per-epoch 12 --disp-batches 4 \
  --num-epochs 1 --checkpoint-every 0 --eval-steps 0 --lr 1e-4 2>&1 | tail -3 | tee -a gpurun_out/r2b.log

echo "== bert b16 s512 fused attn (bf16 bwd) ==" | tee -a gpurun_out/r2b.log
timeout 400 python -m deeplearning_cfn_amd.train --network bert \
  --batch-size 16 --seq-len 512 --steps-per-epoch 8 --disp-batches 4 \
  --num-epochs 1 --checkpoint-every 0 --eval-steps 0 --lr 1e-4 2>&1 | tail -3 | tee -a gpurun_out/r2b.log

echo "== r50 bench (native stem now) b2048 ==" | tee -a gpurun_out/r2b.log
timeout 500 python bench.py --gpus 1 --steps 20 --warmup 8 2>&1 | tail -1 | tee -a gpurun_out/r2b.log

echo "== r50 convergence smoke (memorization) ==" | tee -a gpurun_out/r2b.log
timeout 500 python -m pytest tests/test_convergence_gpu.py -q 2>&1 | tail -2 | tee -a gpurun_out/r2b.log
