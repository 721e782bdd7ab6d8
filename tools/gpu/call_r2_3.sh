set -x
cd /root/repo
mkdir -p gpurun_out
# RPG sweep: more images per row-group => more waves per huffman launch
for rpg in 256 512 1024; do
  PSA_IMAGENET_RPG=$rpg timeout 300 python bench.py --config imagenet --steps 10 --warmup 5 --min-region 4 > gpurun_out/b_rpg$rpg.json 2> gpurun_out/b_rpg$rpg.err
  echo "rpg=$rpg:"; tail -1 gpurun_out/b_rpg$rpg.json | python -c "import json,sys; j=json.load(sys.stdin); print(j['value'], j['timed_region_s'])"
done
# stream-count sweep at the best-looking rpg
for ds in 4 6 8; do
  PSA_IMAGENET_RPG=512 PSA_DECODE_STREAMS=$ds timeout 300 python bench.py --config imagenet --steps 10 --warmup 5 --min-region 4 > gpurun_out/b_ds$ds.json 2> gpurun_out/b_ds$ds.err
  echo "streams=$ds:"; tail -1 gpurun_out/b_ds$ds.json | python -c "import json,sys; j=json.load(sys.stdin); print(j['value'])"
done
# fresh kernel-time profile at current best config
cd /tmp && export TMPDIR=/tmp && cd /root/repo
PSA_IMAGENET_RPG=512 timeout 420 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_r2 -- python bench.py --config imagenet --steps 6 --warmup 4 --min-region 3 > gpurun_out/prof_r2.log 2>&1
grep -A 25 "KERNEL_NAME\|Kernel Name\|NAME" gpurun_out/prof_r2.log | head -40 || true
ls gpurun_out/prof_r2* 2>/dev/null
find gpurun_out/prof_r2 -name "*stats*" | head
