#!/bin/bash
# Round of BN/pooling/wrw validation + bench + profile on a MI355X box.
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

echo "== tr16 probe ==" | tee gpurun_out/perf2.log
timeout 120 ./tools/probe_tr16 > gpurun_out/probe_tr16.txt 2>&1
head -20 gpurun_out/probe_tr16.txt

echo "== pytest gpu (conv + kernels) ==" | tee -a gpurun_out/perf2.log
timeout 900 python -m pytest tests/test_conv_gpu.py tests/test_kernels_gpu.py \
  tests/test_kernels2_gpu.py -x -q 2>&1 | tail -15 | tee -a gpurun_out/perf2.log

echo "== bench native ==" | tee -a gpurun_out/perf2.log
timeout 600 python bench.py --gpus 1 --steps 20 --warmup 8 2>&1 | tail -3 | tee -a gpurun_out/perf2.log

echo "== rocprof stats (native, 5 steps) ==" | tee -a gpurun_out/perf2.log
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
timeout 600 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_p2 -o p2 -- \
  python bench.py --gpus 1 --steps 5 --warmup 5 2>&1 | tail -6 | tee -a gpurun_out/perf2.log
ls gpurun_out/prof_p2/ | tee -a gpurun_out/perf2.log
