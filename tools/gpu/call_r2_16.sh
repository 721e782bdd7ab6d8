cd /root/repo
# A/B on the SAME box: nontemporal coef stores on vs off, interleaved
for trial in 1 2; do
  for nt in 1 0; do
    PSA_JPEG_NT=$nt timeout 300 python bench.py --config imagenet --steps 8 --warmup 4 --min-region 3 > gpurun_out/b_nt${nt}_$trial.json 2>/dev/null
    tail -1 gpurun_out/b_nt${nt}_$trial.json | python -c "import json,sys; print('nt=$nt trial$trial', json.load(sys.stdin)['value'])"
  done
done
