set -x
cd /root/repo
mkdir -p gpurun_out
for combo in "1024 4" "1024 6" "2048 3" "2048 4" "2048 6"; do
  set -- $combo
  PSA_IMAGENET_RPG=$1 PSA_IO_THREADS=$2 timeout 420 python bench.py --config imagenet --steps 10 --warmup 5 --min-region 4 > gpurun_out/b_r$1_i$2.json 2> gpurun_out/b_r$1_i$2.err
  echo "rpg=$1 io=$2:"; tail -1 gpurun_out/b_r$1_i$2.json | python -c "import json,sys; print(json.load(sys.stdin)['value'])" || tail -2 gpurun_out/b_r$1_i$2.err
done
