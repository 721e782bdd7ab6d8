cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 420 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_final -- python bench.py --config imagenet --steps 6 --warmup 4 --min-region 3 > gpurun_out/prof_final.log 2>&1
echo rc=$?
tail -1 gpurun_out/prof_final.log | head -c 200
find gpurun_out/prof_final -name "*.db"
