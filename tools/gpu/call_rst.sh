cd /root/repo
for rst in 1 2; do
  PSA_JPEG_RST_BLOCKS=$rst timeout 500 python bench.py --config imagenet --steps 10 --warmup 5 --min-region 4 > gpurun_out/b_rst$rst.json 2>/dev/null
  tail -1 gpurun_out/b_rst$rst.json | python -c "import json,sys; print('rst=$rst', json.load(sys.stdin)['value'])"
done
