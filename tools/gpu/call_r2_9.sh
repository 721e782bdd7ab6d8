set -x
cd /root/repo
mkdir -p gpurun_out
# validate the new dict-null + decimal GPU paths and everything else
python -m pytest tests -m gpu -x -q 2>&1 | tail -3
# confirm the 551k default config on this box
timeout 420 python bench.py --config imagenet --steps 10 --warmup 5 --min-region 4 > gpurun_out/b_default.json 2> gpurun_out/b_default.err
echo "default:"; tail -1 gpurun_out/b_default.json | python -c "import json,sys; j=json.load(sys.stdin); print(j['value'], 'region', j['timed_region_s'])"
# fresh kernel profile at the 551k config for the huffman-analysis record
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 420 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_r3 -- python bench.py --config imagenet --steps 6 --warmup 4 --min-region 3 > gpurun_out/prof_r3.log 2>&1
echo "prof rc=$?"
find gpurun_out/prof_r3 -name "*.db" | head -2
