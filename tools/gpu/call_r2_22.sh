cd /root/repo
# confirm scalar default
timeout 500 python bench.py --config scalar --steps 10 --warmup 5 --min-region 4 > gpurun_out/b_scdef.json 2>/dev/null
tail -1 gpurun_out/b_scdef.json | python -c "import json,sys; j=json.load(sys.stdin); print('scalar default:', j['value'], 'region', j['timed_region_s'])"
# ngram rowgroup sweep
for rg in 6250 12500 25000; do
  PSA_SEQ_RG=$rg timeout 500 python bench.py --config ngram --steps 10 --warmup 5 --min-region 4 > gpurun_out/b_sq$rg.json 2>/dev/null
  tail -1 gpurun_out/b_sq$rg.json | python -c "import json,sys; print('seq_rg=$rg', json.load(sys.stdin)['value'])" || echo "seq_rg=$rg failed"
done
