#!/bin/bash
# First gpurun call of round 2: validate everything that round 1 could only
# compile-check, and refresh the profile baseline. Run as:
#   /usr/local/graft/bin/gpurun --timeout 1800 -- 'bash tools/round2_gpu_checklist.sh'
set -x
mkdir -p gpurun_out

# 1. full GPU suite on current code
timeout 900 python -m pytest tests -m gpu -x -q 2>&1 | tail -5 | tee gpurun_out/r2_gpu_tests.log

# 2. opt-in conv v3 (BK=64): numerics vs v2/torch + microbench
DISTGPU_CONV_V3=1 timeout 300 python -m pytest tests/test_conv_gpu.py -q 2>&1 | tail -3 | tee gpurun_out/r2_conv_v3.log
DISTGPU_CONV_V3=1 timeout 300 python tools/kernel_bench.py --op conv 2>&1 | tail -20 | tee gpurun_out/r2_conv_v3_bench.log
timeout 300 python tools/kernel_bench.py --op conv 2>&1 | tail -20 | tee gpurun_out/r2_conv_v2_bench.log

# 3. attention + tiles microbench baseline for the round-2 ladder
timeout 300 python tools/kernel_bench.py --op attn 2>&1 | tail -20 | tee gpurun_out/r2_attn_bench.log
timeout 300 python tools/kernel_bench.py --op tiles 2>&1 | tail -10 | tee gpurun_out/r2_tiles_bench.log

# 4. short flagship bench + kernel-level profile
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
timeout 900 python bench.py --steps 3 --warmup 1 2>&1 | tail -2 | tee gpurun_out/r2_bench.log
timeout 900 rocprofv3 --kernel-trace --stats -d gpurun_out/r2_prof -- \
  python bench.py --steps 2 --warmup 1 > gpurun_out/r2_prof_run.log 2>&1 || true
python tools/prof_summary.py gpurun_out/r2_prof gpurun_out/r2_prof_summary.csv || true
