cd /root/repo
for ds in 6 8 10; do
  PSA_DECODE_STREAMS=$ds timeout 420 python bench.py --config imagenet --steps 10 --warmup 5 --min-region 4 > gpurun_out/b_fds$ds.json 2>/dev/null
  tail -1 gpurun_out/b_fds$ds.json | python -c "import json,sys; print('streams=$ds', json.load(sys.stdin)['value'])"
done
