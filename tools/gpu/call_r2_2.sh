set -x
cd /root/repo
mkdir -p gpurun_out
# new GPU tests only (string path, foreign jpeg, nullable byte-array)
python -m pytest tests/test_gpu_decode.py -m gpu -q -k "string or nullable or foreign" 2>&1 | tail -4
# foreign (no-RST) imagenet throughput
PSA_JPEG_RST_BLOCKS=0 timeout 420 python bench.py --config imagenet --steps 10 --warmup 5 --min-region 4 > gpurun_out/b_foreign.json 2> gpurun_out/b_foreign.err
echo "foreign rc=$?"; tail -1 gpurun_out/b_foreign.json
# re-encoded dataset throughput: generate foreign, reencode, then bench on the reencoded dir via the standard config
python - <<'PY'
import os, subprocess, time
os.environ['PSA_BENCH_DATA']='/tmp/psa_bench'
os.makedirs('/tmp/psa_bench', exist_ok=True)
src='/tmp/psa_bench/foreign_src'; dst='/tmp/psa_bench/foreign_rst2'
from petastorm_amd.test_util.dataset_gen import create_imagenet_dataset
os.environ['PSA_JPEG_RST_BLOCKS']='0'
if not os.path.exists(src+'/_SUCCESS'):
    create_imagenet_dataset('file://'+src, num_rows=6144, rows_per_rowgroup=256)
    open(src+'/_SUCCESS','w').write('ok')
del os.environ['PSA_JPEG_RST_BLOCKS']
from petastorm_amd.tools.reencode_dataset import reencode_dataset
t0=time.time()
if not os.path.exists(dst+'/_SUCCESS'):
    rows, cols = reencode_dataset('file://'+src, 'file://'+dst, rst_blocks=2, quality=90)
    open(dst+'/_SUCCESS','w').write('ok')
    print('reencode took %.1fs' % (time.time()-t0))
PY
# measure decode throughput of both dirs with a quick inline loop
python - > gpurun_out/foreign_vs_reencoded.txt 2>&1 <<'PY'
import time, torch, numpy as np
from petastorm_amd import make_batch_reader, TransformSpec, ops
from petastorm_amd.pytorch import BatchedDataLoader
from petastorm_amd.unischema import UnischemaField
ext = ops.ext()
mean = torch.tensor([0.485,0.456,0.406], device='cuda'); inv = 1.0/torch.tensor([0.229,0.224,0.225], device='cuda')
def transform(cols):
    img = cols['image']
    out = torch.empty(img.shape[0],3,img.shape[1],img.shape[2],dtype=torch.float32,device=img.device)
    ext.nhwc_to_nchw_normalize(img,out,mean,inv,1.0/255.0)
    return {'image': out, 'label': cols['label']}
ts = TransformSpec(transform, edit_fields=[UnischemaField('image',np.float32,(3,224,224),None,False)])
for tag, d in [('foreign_noRST','/tmp/psa_bench/foreign_src'), ('reencoded_RST2','/tmp/psa_bench/foreign_rst2')]:
    reader = make_batch_reader('file://'+d, device='cuda', num_epochs=None, shuffle_row_groups=True, seed=1, transform_spec=ts, gpu_options=dict(pipeline_depth=6, decode_streams=4))
    loader = BatchedDataLoader(reader, batch_size=256); it = iter(loader)
    for _ in range(30): next(it)
    torch.cuda.synchronize(); t0=time.perf_counter(); n=0
    while time.perf_counter()-t0 < 5.0:
        b=next(it); n += b['image'].shape[0]
    torch.cuda.synchronize(); el=time.perf_counter()-t0
    print('%s: %.0f samples/s (%d in %.2fs)' % (tag, n/el, n, el))
    reader.stop(); reader.join()
PY
cat gpurun_out/foreign_vs_reencoded.txt
