#!/bin/bash
# Final validation: full GPU suite + short imagenet bench (retry-loop sanity).
set -x
cd "$GRAFT_REPO_ROOT" || exit 1
mkdir -p gpurun_out
timeout 300 python -m pytest tests -m gpu -q -x 2>&1 | tail -4 > gpurun_out/final3_pytest.txt
timeout 150 python bench.py --config imagenet --steps 5 --warmup 2 > gpurun_out/final3_bench.json 2>gpurun_out/final3_bench.log
echo "pytest+bench done"; tail -2 gpurun_out/final3_pytest.txt; tail -c 600 gpurun_out/final3_bench.json
