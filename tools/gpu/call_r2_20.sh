cd /root/repo
for combo in "16 125000" "16 250000" "8 250000" "32 500000" "16 500000"; do
  set -- $combo
  PSA_SCALAR_PAGE_KB=$1 PSA_SCALAR_RG=$2 timeout 420 python bench.py --config scalar --rows 6000000 --steps 10 --warmup 5 --min-region 4 > gpurun_out/b_c$1_$2.json 2>/dev/null
  tail -1 gpurun_out/b_c$1_$2.json | python -c "import json,sys; print('page=$1 rg=$2', json.load(sys.stdin)['value'])" || echo "combo $1/$2 failed"
done
