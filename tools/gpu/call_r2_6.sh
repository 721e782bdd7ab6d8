set -x
cd /root/repo
mkdir -p gpurun_out
for io in 2 3 4 6; do
  PSA_IO_THREADS=$io timeout 300 python bench.py --config imagenet --steps 10 --warmup 5 --min-region 4 > gpurun_out/b_io$io.json 2> gpurun_out/b_io$io.err
  echo "io=$io:"; tail -1 gpurun_out/b_io$io.json | python -c "import json,sys; print(json.load(sys.stdin)['value'])"
done
# also scalar + ngram with more io threads (both are io-heavy)
for cfg in scalar ngram; do
  PSA_IO_THREADS=4 timeout 300 python bench.py --config $cfg --steps 10 --warmup 5 --min-region 4 > gpurun_out/b_io4_$cfg.json 2> gpurun_out/b_io4_$cfg.err
  echo "$cfg io4:"; tail -1 gpurun_out/b_io4_$cfg.json | python -c "import json,sys; print(json.load(sys.stdin)['value'])"
done
