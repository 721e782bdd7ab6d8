set -x
cd /tmp && export TMPDIR=/tmp && cd /root/repo
mkdir -p gpurun_out
# fresh PMC on jpeg_huffman at the tuned (1024rpg, fused) config
timeout 420 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_INSTS_VALU SQ_INSTS_FLAT TCC_HIT TCC_MISS SQ_WAVES --kernel-include-regex jpeg_huffman -d gpurun_out/pmc_r2 -- python bench.py --config imagenet --steps 3 --warmup 2 --min-region 1 > gpurun_out/pmc_r2.log 2>&1
echo "pmc rc=$?"
find gpurun_out/pmc_r2 -type f | head -5
# quick ngram/scalar stream-depth sweep
for ds in 4 8; do
  PSA_DECODE_STREAMS=$ds timeout 300 python bench.py --config ngram --steps 10 --warmup 5 --min-region 3 > gpurun_out/b_ng_ds$ds.json 2>/dev/null
  echo "ngram ds=$ds:"; tail -1 gpurun_out/b_ng_ds$ds.json | python -c "import json,sys; print(json.load(sys.stdin)['value'])"
done
