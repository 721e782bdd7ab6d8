cd /root/repo
python -m pytest tests/test_gpu_decode.py -m gpu -q -k "jpeg or fused or imagenet" 2>&1 | tail -2
for i in 1 2; do
  timeout 420 python bench.py --config imagenet --steps 10 --warmup 5 --min-region 4 > gpurun_out/b_nt$i.json 2>/dev/null
  tail -1 gpurun_out/b_nt$i.json | python -c "import json,sys; print('run'+'$i', json.load(sys.stdin)['value'])"
done
