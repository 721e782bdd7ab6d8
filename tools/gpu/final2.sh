cd /root/repo
# scalar batch probe (x64 default vs x256)
timeout 400 python bench.py --config scalar --batch-size 1024 --steps 10 --warmup 5 --min-region 3 > gpurun_out/b_scb.json 2>/dev/null
tail -1 gpurun_out/b_scb.json | python -c "import json,sys; print('scalar 65k-row batches:', json.load(sys.stdin)['value'])"
# full validation at frozen state
python -m pytest tests -m gpu -q 2>&1 | tail -2
for cfg in imagenet scalar; do
  timeout 500 python bench.py --config $cfg --gpus 1 --steps 20 --warmup 5 > gpurun_out/final2_$cfg.json 2>/dev/null
  tail -1 gpurun_out/final2_$cfg.json | python -c "import json,sys; j=json.load(sys.stdin); print('$cfg', j['value'], 'region', j['timed_region_s'])"
done
python __graft_entry__.py smoke 2>&1 | tail -1
