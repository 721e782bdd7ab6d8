cd /root/repo
for wpk in 64 128 256 512; do
  PSA_WRITER_PAGE_KB=$wpk timeout 500 python bench.py --config ngram --steps 10 --warmup 5 --min-region 4 > gpurun_out/b_w$wpk.json 2>/dev/null
  tail -1 gpurun_out/b_w$wpk.json | python -c "import json,sys; print('writer_page=$wpk', json.load(sys.stdin)['value'])" || echo "wpk=$wpk failed"
done
