set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests -m gpu -q 2>&1 | tail -2
# bench sanity across all four configs after the encoding work
for cfg in imagenet scalar ngram; do
  timeout 420 python bench.py --config $cfg --steps 10 --warmup 5 --min-region 4 > gpurun_out/b3_$cfg.json 2> gpurun_out/b3_$cfg.err
  echo "$cfg:"; tail -1 gpurun_out/b3_$cfg.json | python -c "import json,sys; j=json.load(sys.stdin); print(j['value'], 'region', j['timed_region_s'])" || tail -3 gpurun_out/b3_$cfg.err
done
