set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests -m gpu -x -q 2>&1 | tail -3
timeout 420 python bench.py --config imagenet --steps 10 --warmup 5 --min-region 4 > gpurun_out/b_fused.json 2> gpurun_out/b_fused.err
echo "fused:"; tail -1 gpurun_out/b_fused.json | python -c "import json,sys; j=json.load(sys.stdin); print(j['value'], 'region', j['timed_region_s'])" || tail -5 gpurun_out/b_fused.err
