cd /root/repo
for bs in 256 1024; do
  timeout 500 python bench.py --config ngram --batch-size $bs --steps 10 --warmup 5 --min-region 3 > gpurun_out/b_ngb$bs.json 2>/dev/null
  tail -1 gpurun_out/b_ngb$bs.json | python -c "import json,sys; print('batch-rows', $bs*4, json.load(sys.stdin)['value'])"
done
