#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT" || exit 1
mkdir -p gpurun_out
timeout 180 python examples/imagenet_gpu/main.py > gpurun_out/ex_imagenet_gpu.txt 2>&1
rc=$?
tail -6 gpurun_out/ex_imagenet_gpu.txt
exit $rc
