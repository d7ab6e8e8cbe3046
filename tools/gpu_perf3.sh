#!/bin/bash
# wrw-fix validation + per-shape conv table + PMC counters on bench.
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

echo "== pytest conv gpu ==" | tee gpurun_out/perf3.log
timeout 600 python -m pytest tests/test_conv_gpu.py -x -q 2>&1 | tail -8 | tee -a gpurun_out/perf3.log

echo "== per-shape conv table ==" | tee -a gpurun_out/perf3.log
timeout 900 python tools/bench_conv_shapes.py --batch 256 \
  > gpurun_out/conv_shapes.txt 2>&1
tail -8 gpurun_out/conv_shapes.txt | tee -a gpurun_out/perf3.log

echo "== bench native ==" | tee -a gpurun_out/perf3.log
timeout 600 python bench.py --gpus 1 --steps 20 --warmup 8 2>&1 | tail -2 | tee -a gpurun_out/perf3.log

echo "== PMC on conv kernels (3 steps) ==" | tee -a gpurun_out/perf3.log
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
timeout 900 rocprofv3 --kernel-trace --stats \
  --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_LDS_BANK_CONFLICT \
  -d gpurun_out/prof_pmc -o pmc -- \
  python bench.py --gpus 1 --steps 3 --warmup 3 --batch-size 128 2>&1 | tail -4 | tee -a gpurun_out/perf3.log
ls gpurun_out/prof_pmc/ | tee -a gpurun_out/perf3.log
