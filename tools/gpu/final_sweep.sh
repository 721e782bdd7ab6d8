set -x
cd /root/repo
mkdir -p gpurun_out
python -m pytest tests -m gpu -q 2>&1 | tail -2
python __graft_entry__.py smoke 2>&1 | tail -2 || python -c "import __graft_entry__; __graft_entry__.smoke()" 2>&1 | tail -2
# driver-exact invocations
for cfg in imagenet scalar ngram helloworld; do
  timeout 600 python bench.py --config $cfg --gpus 1 --steps 20 --warmup 5 > gpurun_out/final_$cfg.json 2> gpurun_out/final_$cfg.err
  echo "$cfg rc=$?:"; tail -1 gpurun_out/final_$cfg.json | python -c "import json,sys; j=json.load(sys.stdin); print(j['value'], 'region', j['timed_region_s'], 'bps', j['config'].get('batches_per_step'))" || tail -3 gpurun_out/final_$cfg.err
done
# 2-rank self-spawn (shared GPU -> gloo fallback)
timeout 500 python bench.py --gpus 2 --config imagenet --steps 8 --warmup 4 --min-region 3 > gpurun_out/final_2rank.json 2> gpurun_out/final_2rank.err
echo "2rank rc=$?:"; tail -1 gpurun_out/final_2rank.json | python -c "import json,sys; print(json.load(sys.stdin)['value'])" || tail -3 gpurun_out/final_2rank.err
# stability probe across five pipelines
timeout 420 python tools/stability_probe.py --minutes 4 > gpurun_out/final_stability.txt 2>&1
echo "probe rc=$?"; tail -8 gpurun_out/final_stability.txt
