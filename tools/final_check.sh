#!/bin/bash
# Final round-2 validation sweep: full GPU suite + smoke + all benches.
set -x
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -5 | tee gpurun_out/final_pytest.log
timeout 240 python -c "import __graft_entry__ as ge; ge.smoke(); print('smoke ok')" 2>&1 | tail -3 | tee gpurun_out/final_smoke.log
timeout 420 python bench.py 2>&1 | tail -2 | tee gpurun_out/final_bench.log
timeout 420 python tools/bench_infer.py 2>&1 | tail -6 | tee gpurun_out/final_infer.log
timeout 600 python tools/bench_suite.py 3 4 5 2>&1 | tail -4 | tee gpurun_out/final_suite.log
echo DONE_ALL
