cd /root/repo
# more row-groups at rpg=1024: deeper overlap, same launch shape
for rows in 24576 49152; do
  timeout 600 python bench.py --config imagenet --rows $rows --steps 10 --warmup 5 --min-region 4 > gpurun_out/b_rows$rows.json 2>/dev/null
  tail -1 gpurun_out/b_rows$rows.json | python -c "import json,sys; print('rows=$rows', json.load(sys.stdin)['value'])"
done
# depth sweep at the default dataset
for depth in 6 8 10; do
  PSA_PIPELINE_DEPTH=$depth timeout 420 python bench.py --config imagenet --steps 10 --warmup 5 --min-region 4 > gpurun_out/b_d$depth.json 2>/dev/null
  tail -1 gpurun_out/b_d$depth.json | python -c "import json,sys; print('depth=$depth', json.load(sys.stdin)['value'])"
done
