cd /root/repo
PSA_IMAGENET_RPG=2048 timeout 900 python bench.py --config imagenet --rows 98304 --steps 10 --warmup 5 --min-region 4 > gpurun_out/b_2048x48.json 2> gpurun_out/b_2048x48.err
tail -1 gpurun_out/b_2048x48.json | python -c "import json,sys; print('2048x48:', json.load(sys.stdin)['value'])" || tail -3 gpurun_out/b_2048x48.err
# same box reference point at current default
timeout 420 python bench.py --config imagenet --steps 10 --warmup 5 --min-region 4 > gpurun_out/b_defref.json 2>/dev/null
tail -1 gpurun_out/b_defref.json | python -c "import json,sys; print('default:', json.load(sys.stdin)['value'])"
