#!/bin/bash
# Round-2 batch C: igemm hoisting effect on conv shapes + attention micro.
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

echo "== conv parity ==" | tee gpurun_out/r2c.log
timeout 600 python -m pytest tests/test_conv_gpu.py tests/test_gemm_gpu.py -q 2>&1 | tail -3 | tee -a gpurun_out/r2c.log

echo "== conv shapes b1024 (post-hoist) ==" | tee -a gpurun_out/r2c.log
timeout 600 python tools/bench_conv_shapes.py --batch 1024 > gpurun_out/conv_shapes_b1024_r2c.txt 2>&1
tail -6 gpurun_out/conv_shapes_b1024_r2c.txt | tee -a gpurun_out/r2c.log

echo "== attention kernel micro ==" | tee -a gpurun_out/r2c.log
timeout 300 python - <<'EOF' 2>&1 | tee -a gpurun_out/r2c.log
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
    def bmm():
        s = (q @ k.transpose(-1,-2)) * sc
        p = torch.softmax(s.float(), dim=-1).to(q.dtype)
        return p @ v
    print(f"B{B} H{H} S{S}: fused={t(lambda: n.attn_fwd(q,k,v,sc)):.3f} ms  "
          f"bmm+softmax={t(bmm):.3f} ms")
EOF

echo "== r50 bench b2048 (post-hoist) ==" | tee -a gpurun_out/r2c.log
timeout 500 python bench.py --gpus 1 --steps 20 --warmup 8 2>&1 | tail -1 | tee -a gpurun_out/r2c.log
echo "== r50 bench b256 ==" | tee -a gpurun_out/r2c.log
timeout 500 python bench.py --gpus 1 --steps 20 --warmup 8 --batch-size 256 2>&1 | tail -1 | tee -a gpurun_out/r2c.log
