cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 420 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_INSTS_VALU TCC_HIT TCC_MISS SQ_WAVES --kernel-include-regex "jpeg_color_norm|snappy_decompress" -d gpurun_out/pmc_new -- python bench.py --config imagenet --steps 4 --warmup 3 --min-region 2 > gpurun_out/pmc_new.log 2>&1
echo "rc=$?"
timeout 300 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_INSTS_VALU TCC_HIT TCC_MISS SQ_WAVES --kernel-include-regex "snappy_decompress|plain_fixed" -d gpurun_out/pmc_sc -- python bench.py --config scalar --steps 4 --warmup 3 --min-region 2 > gpurun_out/pmc_sc.log 2>&1
echo "rc=$?"
find gpurun_out/pmc_new gpurun_out/pmc_sc -name "*.db"
