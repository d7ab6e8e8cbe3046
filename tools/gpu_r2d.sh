#!/bin/bash
# Round-2 batch D: fused attention backward validation + BERT perf,
# graph test, pipe evidence, short-K 8ph threshold probe.
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

echo "== attention tests (incl fused bwd) ==" | tee gpurun_out/r2d.log
timeout 600 python -m pytest tests/test_attention_gpu.py tests/test_graph_gpu.py -q 2>&1 | tail -4 | tee -a gpurun_out/r2d.log

echo "== attn bwd micro ==" | tee -a gpurun_out/r2d.log
timeout 300 python - <<'EOF' 2>&1 | tee -a gpurun_out/r2d.log
import torch, time, sys
sys.path.insert(0, ".")
from deeplearning_cfn_amd.ops import native
n = native()
def t(fn, iters=20, warm=5):
    for _ in range(warm): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/iters*1e3
for B,H,S in [(32,12,128),(16,12,512)]:
    q = torch.randn(B,H,S,64,device="cuda").bfloat16()
    k, v = torch.randn_like(q), torch.randn_like(q)
    sc = 0.125
    o, lse = n.attn_fwd(q,k,v,sc)
    g = torch.randn_like(o)
    def compose_bwd():
        s = (q @ k.transpose(-1,-2)).float() * sc
        p = torch.exp(s - lse.unsqueeze(-1))
        dv = p.to(q.dtype).transpose(-1,-2) @ g
        dp = (g @ v.transpose(-1,-2)).float()
        delta = (g.float()*o.float()).sum(-1,keepdim=True)
        ds = (p*(dp-delta)*sc).to(q.dtype)
        return ds @ k, ds.transpose(-1,-2) @ q, dv
    print(f"B{B} H{H} S{S}: fused_bwd={t(lambda: n.attn_bwd(g,q,k,v,o,lse,sc)):.3f} ms  "
          f"compose_bwd={t(compose_bwd):.3f} ms")
EOF

echo "== bert b32 s128 / b16 s512 (fused fwd+bwd) ==" | tee -a gpurun_out/r2d.log
timeout 400 python -m deeplearning_cfn_amd.train --network bert \
  --batch-size 32 --seq-len 128 --steps-per-epoch 12 --disp-batches 4 \
  --num-epochs 1 --checkpoint-every 0 --eval-steps 0 --lr 1e-4 2>&1 | tail -3 | tee -a gpurun_out/r2d.log
timeout 400 python -m deeplearning_cfn_amd.train --network bert \
  --batch-size 16 --seq-len 512 --steps-per-epoch 8 --disp-batches 4 \
  --num-epochs 1 --checkpoint-every 0 --eval-steps 0 --lr 1e-4 2>&1 | tail -3 | tee -a gpurun_out/r2d.log

echo "== pipe throughput evidence ==" | tee -a gpurun_out/r2d.log
timeout 300 python tools/bench_pipe.py 2>&1 | tee -a gpurun_out/r2d.log

echo "== short-K 8ph threshold probe (dgrad K=256 rows) ==" | tee -a gpurun_out/r2d.log
for MK in 384 256 128; do
  echo "-- DLX_8PH_MINK=$MK --" | tee -a gpurun_out/r2d.log
  DLX_8PH_MINK=$MK timeout 300 python - <<'EOF' 2>&1 | tee -a gpurun_out/r2d.log
import torch, time, sys
sys.path.insert(0, ".")
from deeplearning_cfn_amd.ops import native
n = native()
def t(fn, iters=10, warm=3):
    for _ in range(warm): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/iters*1e3
b = 1024
for ci,h,co in [(512,28,256),(256,56,64),(1024,14,256),(2048,7,512)]:
    # dgrad of 1x1/s1: plain GEMM M=b*h*h, N=ci, K=co
    dy = torch.randn(b,co,h,h,device="cuda").bfloat16().contiguous(memory_format=torch.channels_last)
    w = (torch.randn(co,ci,1,1,device="cuda")*0.05).bfloat16()
    print(f"dgrad {ci}x{h}^2<-{co}: {t(lambda: n.conv2d_dgrad(dy,w,h,h,1,0)):.3f} ms")
EOF
done
