set -x
cd /root/repo
mkdir -p gpurun_out
# cProfile the consumer-side dispatch of a short imagenet run
timeout 420 python -m cProfile -o gpurun_out/bench.prof bench.py --config imagenet --steps 5 --warmup 3 --min-region 2 > gpurun_out/b_prof.json 2>&1
python - <<'PY' > gpurun_out/prof_top.txt 2>&1
import pstats
p = pstats.Stats('gpurun_out/bench.prof')
p.sort_stats('cumulative').print_stats(30)
p.sort_stats('tottime').print_stats(30)
PY
tail -1 gpurun_out/b_prof.json | head -c 400; echo
grep -A 40 "Ordered by: internal time" gpurun_out/prof_top.txt | head -45
# rpg 1024 + io3 combo
PSA_IMAGENET_RPG=1024 timeout 300 python bench.py --config imagenet --steps 10 --warmup 5 --min-region 4 > gpurun_out/b_1024io3.json 2>/dev/null
echo "rpg1024 io3:"; tail -1 gpurun_out/b_1024io3.json | python -c "import json,sys; print(json.load(sys.stdin)['value'])"
