#!/bin/bash
# HEAD validation on a MI355X box: gpu tests, bench (native + torch), profile.
# Run via: /usr/local/graft/bin/gpurun --timeout 1500 -- 'bash tools/gpu_validate.sh'
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

echo "== pytest -m gpu ==" | tee gpurun_out/validate.log
timeout 900 python -m pytest tests -m gpu -x -q 2>&1 | tail -25 | tee -a gpurun_out/validate.log

echo "== smoke ==" | tee -a gpurun_out/validate.log
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" 2>&1 | tail -5 | tee -a gpurun_out/validate.log

echo "== bench native ==" | tee -a gpurun_out/validate.log
timeout 600 python bench.py --gpus 1 --steps 20 --warmup 8 2>&1 | tail -3 | tee -a gpurun_out/validate.log

echo "== bench torch impl ==" | tee -a gpurun_out/validate.log
timeout 600 env DLX_IMPL=torch python bench.py --gpus 1 --steps 20 --warmup 8 --impl torch 2>&1 | tail -3 | tee -a gpurun_out/validate.log

echo "== rocprof stats (native, 5 steps) ==" | tee -a gpurun_out/validate.log
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
timeout 600 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_v -o vrun -- \
  python bench.py --gpus 1 --steps 5 --warmup 5 2>&1 | tail -40 | tee -a gpurun_out/validate.log
ls -la gpurun_out/prof_v/ | tee -a gpurun_out/validate.log
