set -x
cd /root/repo
python -m pytest tests -m gpu -x -q 2>&1 | tail -5
mkdir -p gpurun_out
# RCCL world=1: real nccl/RCCL init + epoch collectives on the 1-GPU lease
timeout 420 python -m torch.distributed.run --nnodes=1 --nproc-per-node 1 --master-addr 127.0.0.1 --master-port 29871 bench.py --config imagenet --steps 20 --warmup 5 > gpurun_out/b_imagenet_rccl.json 2> gpurun_out/b_imagenet_rccl.err
echo "rccl world1 rc=$?"; tail -2 gpurun_out/b_imagenet_rccl.err; tail -1 gpurun_out/b_imagenet_rccl.json
# plain single-process runs with the new 5s default region
timeout 420 python bench.py --config scalar --steps 20 --warmup 5 > gpurun_out/b_scalar.json 2> gpurun_out/b_scalar.err
echo "scalar rc=$?"; tail -1 gpurun_out/b_scalar.json
# interop GPU test explicitly (already in -m gpu but show verbose)
python -m pytest tests/test_reference_interop.py -m gpu -q 2>&1 | tail -3
