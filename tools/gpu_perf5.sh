#!/bin/bash
# Validate maxpool fix + stride-2 dgrad; collect SUMMARIZED profiles only.
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

echo "== pytest gpu full ==" | tee gpurun_out/perf5.log
timeout 900 python -m pytest tests -m gpu -x -q 2>&1 | tail -6 | tee -a gpurun_out/perf5.log

echo "== per-shape conv (s2 dgrad focus) ==" | tee -a gpurun_out/perf5.log
timeout 900 python tools/bench_conv_shapes.py --batch 256 \
  > gpurun_out/conv_shapes5.txt 2>&1
tail -6 gpurun_out/conv_shapes5.txt | tee -a gpurun_out/perf5.log

echo "== bench native ==" | tee -a gpurun_out/perf5.log
timeout 600 python bench.py --gpus 1 --steps 20 --warmup 8 2>&1 | tail -2 | tee -a gpurun_out/perf5.log

echo "== stats profile (5 steps, summarized) ==" | tee -a gpurun_out/perf5.log
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
timeout 600 rocprofv3 --kernel-trace --stats -d /tmp/prof_p5 -o p5 -- \
  python bench.py --gpus 1 --steps 5 --warmup 5 2>&1 | tail -2 | tee -a gpurun_out/perf5.log
python tools/prof_summary.py /tmp/prof_p5/p5_results.db \
  gpurun_out/r50_step_kernels_p5.md "ResNet-50 b256 bf16 native, 5+5 steps (post BN-fusion + s2-dgrad)" \
  | tee -a gpurun_out/perf5.log

echo "== PMC igemm (summarized) ==" | tee -a gpurun_out/perf5.log
timeout 900 rocprofv3 --kernel-trace \
  --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_LDS_BANK_CONFLICT \
  -d /tmp/prof_pmc5 -o pmc5 -- \
  python tools/bench_conv_shapes.py --batch 256 > /tmp/shapes_pmc.txt 2>&1
python tools/pmc_summary.py /tmp/prof_pmc5/pmc5_results.db \
  gpurun_out/igemm_pmc_p5.md "igemm PMC over ResNet-50 conv shapes b256" \
  | tee -a gpurun_out/perf5.log
du -sh gpurun_out | tee -a gpurun_out/perf5.log
