set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests -m gpu -x -q 2>&1 | tail -3
# 2-rank shared-GPU validation (gloo) with the new defaults
timeout 420 python bench.py --gpus 2 --config imagenet --steps 8 --warmup 4 --min-region 3 > gpurun_out/b_2rank.json 2> gpurun_out/b_2rank.err
echo "2rank rc=$?"; tail -1 gpurun_out/b_2rank.json; tail -3 gpurun_out/b_2rank.err
# stability probe incl. zstd/lz4 variants
timeout 500 python tools/stability_probe.py --minutes 5 > gpurun_out/stability_r2.txt 2>&1
echo "probe rc=$?"; cat gpurun_out/stability_r2.txt
# stage-times / dispatch-cost capture
PSA_TIMING=1 timeout 300 python bench.py --config imagenet --steps 6 --warmup 4 --min-region 3 > gpurun_out/b_timing.json 2> gpurun_out/b_timing.err
grep -h "stage_times\|staging" gpurun_out/b_timing.err | head -5
