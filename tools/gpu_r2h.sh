#!/bin/bash
# Round-2 final validation: full GPU suite, smoke, benches, rocprof
# step profiles (r50 + BERT) for profiles/.
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

echo "== full GPU pytest ==" | tee gpurun_out/r2h.log
timeout 1200 python -m pytest tests -m gpu -q -rs 2>&1 | tail -8 | tee -a gpurun_out/r2h.log

echo "== smoke ==" | tee -a gpurun_out/r2h.log
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" 2>&1 | tail -2 | tee -a gpurun_out/r2h.log

echo "== bench b2048 / b256 ==" | tee -a gpurun_out/r2h.log
timeout 500 python bench.py --gpus 1 --steps 20 --warmup 8 2>&1 | tail -1 | tee -a gpurun_out/r2h.log
timeout 400 python bench.py --gpus 1 --steps 20 --warmup 8 --batch-size 256 2>&1 | tail -1 | tee -a gpurun_out/r2h.log

echo "== rocprof r50 step ==" | tee -a gpurun_out/r2h.log
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
timeout 500 rocprofv3 --kernel-trace --stats -d /tmp/prof_r2 -o r2 -- \
  python bench.py --gpus 1 --steps 5 --warmup 5 --batch-size 1024 2>&1 | tail -1 | tee -a gpurun_out/r2h.log
python tools/prof_summary.py /tmp/prof_r2/r2_results.db \
  gpurun_out/r50_step_kernels_r2.md \
  "ResNet-50 b1024 bf16 native step, round 2 (native stem, hoisted igemm, deterministic wrw), 5+5 steps" \
  | head -5 | tee -a gpurun_out/r2h.log

echo "== rocprof BERT step ==" | tee -a gpurun_out/r2h.log
timeout 500 rocprofv3 --kernel-trace --stats -d /tmp/prof_bert -o bt -- \
  python -m deeplearning_cfn_amd.train --network bert --batch-size 32 \
  --seq-len 128 --steps-per-epoch 6 --disp-batches 6 --num-epochs 1 \
  --checkpoint-every 0 --eval-steps 0 --lr 1e-4 2>&1 | tail -1 | tee -a gpurun_out/r2h.log
python tools/prof_summary.py /tmp/prof_bert/bt_results.db \
  gpurun_out/bert_step_kernels_r2.md \
  "BERT-base b32 s128 bf16 step, round 2 (fused flash attention fwd+bwd, MFMA MLM head), 6 steps" \
  | head -5 | tee -a gpurun_out/r2h.log
grep -c "Cijk" gpurun_out/bert_step_kernels_r2.md | tee -a gpurun_out/r2h.log || true
