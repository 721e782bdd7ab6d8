cd /root/repo
for pk in 16 32 64 128; do
  PSA_SCALAR_PAGE_KB=$pk timeout 420 python bench.py --config scalar --steps 10 --warmup 5 --min-region 4 > gpurun_out/b_pk$pk.json 2>/dev/null
  tail -1 gpurun_out/b_pk$pk.json | python -c "import json,sys; print('page=${pk}KiB', json.load(sys.stdin)['value'])"
done
# rowgroup-size sweep at the winning-looking page size
for rg in 62500 125000 250000; do
  PSA_SCALAR_PAGE_KB=32 PSA_SCALAR_RG=$rg timeout 420 python bench.py --config scalar --steps 10 --warmup 5 --min-region 4 > gpurun_out/b_rg$rg.json 2>/dev/null
  tail -1 gpurun_out/b_rg$rg.json | python -c "import json,sys; print('rg=$rg', json.load(sys.stdin)['value'])" || echo "rg=$rg failed"
done
