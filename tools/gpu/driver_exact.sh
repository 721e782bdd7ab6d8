#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT" || exit 1
mkdir -p gpurun_out
timeout 170 python bench.py --gpus 1 > gpurun_out/driver_exact.json 2>gpurun_out/driver_exact.log
rc=$?
tail -c 700 gpurun_out/driver_exact.json; echo; tail -2 gpurun_out/driver_exact.log
exit $rc
