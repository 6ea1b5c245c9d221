#!/bin/bash
# One-command local verification (CPU container): what the driver checks,
# runnable any time. GPU-side equivalent: tools/round2_gpu_checklist.sh.
set -e
cd "$(dirname "$0")/.."
echo "== build (hipcc gfx950 cross-compile) =="
python -c "import __graft_entry__ as g; g.build()" | tail -1
echo "== CPU suite =="
python -m pytest tests/ -q -m "not gpu" | tail -1
echo "== gpu-marked collection (import check) =="
python -m pytest tests/ --collect-only -q -m gpu | tail -1
echo "== bench contract (tiny) =="
python -m pytest tests/test_bench_contract.py -q | tail -1
echo "== all checks passed =="
