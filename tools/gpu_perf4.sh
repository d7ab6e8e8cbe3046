#!/bin/bash
# Validate BN-residual fusion + clean bench + PMC on igemm kernels.
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

echo "== pytest gpu full ==" | tee gpurun_out/perf4.log
timeout 900 python -m pytest tests -m gpu -x -q 2>&1 | tail -6 | tee -a gpurun_out/perf4.log

echo "== bench native ==" | tee -a gpurun_out/perf4.log
timeout 600 python bench.py --gpus 1 --steps 20 --warmup 8 2>&1 | tail -2 | tee -a gpurun_out/perf4.log

echo "== stats profile (5 steps) ==" | tee -a gpurun_out/perf4.log
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
timeout 600 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_p4 -o p4 -- \
  python bench.py --gpus 1 --steps 5 --warmup 5 2>&1 | tail -3 | tee -a gpurun_out/perf4.log

echo "== PMC on conv shapes (igemm focus) ==" | tee -a gpurun_out/perf4.log
timeout 900 rocprofv3 --kernel-trace \
  --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_LDS_BANK_CONFLICT \
  -d gpurun_out/prof_pmc4 -o pmc4 -- \
  python tools/bench_conv_shapes.py --batch 256 > gpurun_out/conv_shapes_pmc.txt 2>&1
tail -6 gpurun_out/conv_shapes_pmc.txt | tee -a gpurun_out/perf4.log
ls gpurun_out/prof_pmc4/ | tee -a gpurun_out/perf4.log
