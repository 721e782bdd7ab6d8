#!/usr/bin/env python3
"""Flagship benchmark: samples/sec through make_batch_reader -> PyTorch
loader on the BASELINE.json configs.

Default config reproduces the reference's headline metric class
(BASELINE.md): an ImageNet-style petastorm dataset (224x224x3 jpeg
CompressedImageCodec + int32 label) read through
``make_batch_reader(device='cuda')`` (on-GPU snappy/page/jpeg decode +
fused NHWC->NCHW normalize) into batches.  A "step" is one batch of
``--batch-size`` samples per GPU.

Multi-GPU: one rank per GPU via torch.distributed.run; sharding is
``cur_shard=rank / shard_count=world`` with the RCCL epoch broadcast/
all-gather (petastorm_amd.parallel.epochs).  The printed ``value`` is the
whole-job samples/sec (all ranks).

The reference's published numbers for this metric are 709.84 samples/sec
(defaults) and 653.10 (long run) on unspecified CPU hardware
(docs/benchmarks_tutorial.rst:20-37) -> vs_baseline = value / 709.84.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import numpy as np  # noqa: E402
import torch  # noqa: E402

BASELINE_SAMPLES_PER_SEC = 709.84


def _dist_env():
    rank = int(os.environ.get('RANK', '0'))
    world = int(os.environ.get('WORLD_SIZE', '1'))
    local = int(os.environ.get('LOCAL_RANK', str(rank)))
    return rank, world, local


def _init_dist(world):
    import torch.distributed as dist
    if dist.is_initialized():
        return None
    # Under torchrun, init the process group even at world=1 so the real
    # backend (RCCL on a GPU box) initializes and the epoch collectives
    # run over it — keeps the 8-GPU path exercised on 1-GPU leases.
    if world <= 1 and 'WORLD_SIZE' not in os.environ:
        return None
    kwargs = {}
    backend = os.environ.get('PSA_DIST_BACKEND')
    if backend is None:
        # nccl (=RCCL on ROCm) whenever every rank can own a GPU; gloo when
        # ranks must share one GPU (single-GPU validation boxes: two RCCL
        # ranks on one device deadlock)
        if torch.cuda.is_available() and \
                torch.cuda.device_count() >= world:
            backend = 'nccl'
        else:
            backend = 'gloo'
    if backend == 'nccl':
        local = int(os.environ.get('LOCAL_RANK', '0'))
        local = min(local, torch.cuda.device_count() - 1)
        kwargs['device_id'] = torch.device('cuda', local)
    dist.init_process_group(backend=backend, **kwargs)
    return dist


def _self_spawn(args):
    """The driver may invoke ``bench.py --gpus N`` directly (no torchrun);
    spawn one rank per GPU ourselves via torch.distributed.run."""
    import socket
    import subprocess
    with socket.socket() as s:
        s.bind(('127.0.0.1', 0))
        port = s.getsockname()[1]
    cmd = [sys.executable, '-m', 'torch.distributed.run',
           '--nnodes=1', '--nproc-per-node', str(args.gpus),
           '--master-addr', '127.0.0.1', '--master-port', str(port),
           os.path.abspath(__file__)] + sys.argv[1:]
    env = dict(os.environ)
    env.setdefault('HSA_ENABLE_IPC_MODE_LEGACY', '0')
    return subprocess.call(cmd, env=env)


def _sync(device):
    if torch.cuda.is_available():
        torch.cuda.synchronize(device)


def _barrier(dist):
    if dist is not None:
        dist.barrier()


def _dataset_dir(tag, rank, dist, gen_fn):
    """Rank 0 generates the synthetic dataset; everyone reads it."""
    base = os.environ.get('PSA_BENCH_DATA', '/tmp/psa_bench')
    path = os.path.join(base, tag)
    marker = os.path.join(path, '_SUCCESS')
    if rank == 0 and not os.path.exists(marker):
        os.makedirs(path, exist_ok=True)
        gen_fn('file://' + path)
        open(marker, 'w').write('ok')
    _barrier(dist)
    return 'file://' + path


def bench_imagenet(args, rank, world, device, dist):
    """BASELINE configs 3/4: jpeg CompressedImageCodec + on-GPU decode +
    NHWC->NCHW normalize TransformSpec."""
    from petastorm_amd import make_batch_reader
    from petastorm_amd import ops
    from petastorm_amd.pytorch import BatchedDataLoader
    from petastorm_amd.test_util.dataset_gen import create_imagenet_dataset

    # >= 8 row-groups are required so every rank of an 8-GPU run gets data
    # (sharding is per row-group, reference reader.py:573-597).
    # 24 uniform row-groups -> every one of 8 shards gets exactly 3.
    # rpg x io_threads sweep on MI355X (profiles/RESULTS.md r2.4):
    # 256=228k, 512=404k, 1024=551k, 2048=385k samples/s
    rpg = int(os.environ.get('PSA_IMAGENET_RPG', '1024'))
    n_rows = args.rows or 24 * rpg  # 24 uniform row-groups (equal 8-GPU shards)
    rst = os.environ.get('PSA_JPEG_RST_BLOCKS', '2')
    url = _dataset_dir('imagenet_{}_r{}_g{}'.format(n_rows, rst, rpg), rank,
                       dist,
                       lambda u: create_imagenet_dataset(
                           u, num_rows=n_rows, rows_per_rowgroup=rpg))

    ops.ext()  # loud failure if the native extension is missing
    # fused into the jpeg color kernel epilogue on the GPU route
    # (transform.FusedImageNormalize): no NHWC uint8 intermediate, no
    # separate normalize kernel
    from petastorm_amd.transform import fused_image_normalize
    ts = fused_image_normalize('image', mean=[0.485, 0.456, 0.406],
                               std=[0.229, 0.224, 0.225])
    reader = make_batch_reader(
        url, device=str(device), num_epochs=None, shuffle_row_groups=True,
        seed=1234, transform_spec=ts,
        cur_shard=rank if world > 1 else None,
        shard_count=world if world > 1 else None,
        gpu_options=dict(
            pipeline_depth=int(os.environ.get('PSA_PIPELINE_DEPTH', '6')),
            decode_streams=int(os.environ.get('PSA_DECODE_STREAMS', '6')),
            io_threads=int(os.environ.get('PSA_IO_THREADS', '4'))))
    loader = BatchedDataLoader(reader, batch_size=args.batch_size)

    it = iter(loader)

    def step():
        b = next(it)
        return b['image'].shape[0]

    result = _run_timed(args, step, device, dist, world)
    if dist is not None and hasattr(reader, 'epoch_stats'):
        # RCCL epoch all-gather over xGMI (BASELINE config 4): every rank
        # reports its consumed rows at this synchronized point
        stats = reader.epoch_stats()
        if rank == 0:
            print('epoch_stats (rows/rank):', stats, file=sys.stderr)
    reader.stop()
    reader.join()
    if rank == 0 and reader.diagnostics.get('cpu_assist_columns'):
        print('WARNING: cpu-assist columns: {}'.format(
            reader.diagnostics['cpu_assist_columns']), file=sys.stderr)
    if os.environ.get('PSA_TIMING') == '1' and rank == 0:
        diag = reader.diagnostics
        print('stage_times:', diag.get('stage_times'),
              'staging_allocs:', diag.get('staging_allocs'),
              'staging_copy_s:', diag.get('staging_copy_s'),
              'by_key:', diag.get('staging_copy_by_key'),
              file=sys.stderr)
    return result, {
        'model': 'ImageNetSchema(224x224x3 jpeg CompressedImageCodec + '
                 'int32 label)',
        'global_batch': args.batch_size * world,
        'seq_len': None,
        'parallelism': 'dp{}'.format(world),
        'pipeline': 'make_batch_reader(device=cuda): GPU page+jpeg decode + '
                    'fused NHWC->NCHW normalize',
    }


def bench_scalar(args, rank, world, device, dist):
    """BASELINE config 2: scalar-only Parquet, on-GPU snappy column decode."""
    from petastorm_amd import make_batch_reader
    from petastorm_amd.pytorch import BatchedDataLoader
    from petastorm_amd.test_util.dataset_gen import create_scalar_dataset

    # rowgroup/page sweep on MI355X (profiles/RESULTS.md r2.7): the
    # config was host-dispatch-bound at 62.5k-row groups; 1M-row groups
    # with 16 KiB pages reach ~206M rows/s (24 groups -> 3 per shard at
    # 8 GPUs)
    n_rows = args.rows or 24_000_000
    comp = os.environ.get('PSA_SCALAR_COMPRESSION', 'snappy')
    page_kb = os.environ.get('PSA_SCALAR_PAGE_KB', '16')
    rg_rows = int(os.environ.get('PSA_SCALAR_RG', '1000000'))
    url = _dataset_dir('scalar_{}_{}_p{}_g{}'.format(n_rows, comp, page_kb,
                                                     rg_rows),
                       rank, dist,
                       lambda u: create_scalar_dataset(
                           u, num_rows=n_rows, rowgroup_size=rg_rows,
                           compression=comp))
    reader = make_batch_reader(
        url, device=str(device), num_epochs=None, shuffle_row_groups=True,
        seed=7, schema_fields=['id', 'f0', 'f1', 'f2', 'f3', 'i0', 'i1'],
        cur_shard=rank if world > 1 else None,
        shard_count=world if world > 1 else None,
        gpu_options=dict(pipeline_depth=6))
    loader = BatchedDataLoader(reader, batch_size=args.batch_size * 64)
    it = iter(loader)

    def step():
        b = next(it)
        return b['id'].shape[0]

    result = _run_timed(args, step, device, dist, world)
    reader.stop()
    reader.join()
    if os.environ.get('PSA_TIMING') == '1' and rank == 0:
        print('stage_times:', reader.diagnostics.get('stage_times'),
              file=sys.stderr)
    return result, {
        'model': 'scalar-parquet (8 float64 + 8 int64 cols, snappy)',
        'global_batch': args.batch_size * 64 * world,
        'seq_len': None,
        'parallelism': 'dp{}'.format(world),
        'pipeline': 'make_batch_reader(device=cuda): GPU snappy + PLAIN '
                    'decode, pinned H2D',
    }


def bench_helloworld_cpu(args, rank, world, device, dist):
    """BASELINE config 1: HelloWorld via make_reader on the CPU thread pool
    (plumbing parity benchmark — the exact reference headline config)."""
    from petastorm_amd import make_reader
    from petastorm_amd.test_util.dataset_gen import create_hello_world_dataset

    n_rows = args.rows or 400
    url = _dataset_dir('hello_{}'.format(n_rows), rank, dist,
                       lambda u: create_hello_world_dataset(
                           u, num_rows=n_rows, rowgroup_size_mb=4))
    reader = make_reader(url, reader_pool_type='thread', workers_count=3,
                         num_epochs=None, shuffle_row_groups=True)
    it = iter(reader)

    def step():
        # the reference benchmark counts single next(reader) rows
        # (petastorm/benchmark/throughput.py:68-90); a "step" here is
        # batch_size of them so step timing stays comparable
        for _ in range(args.batch_size):
            next(it)
        return args.batch_size

    result = _run_timed(args, step, device, dist, world)
    reader.stop()
    reader.join()
    return result, {
        'model': 'HelloWorldSchema (int32 id + 128x256x3 png + 4d uint8 '
                 'ndarray)',
        'global_batch': args.batch_size * world,
        'seq_len': None,
        'parallelism': 'cpu-threadpool x{}'.format(world),
        'pipeline': 'make_reader, thread pool, 3 workers (reference headline '
                    'config, docs/benchmarks_tutorial.rst)',
    }


def bench_ngram(args, rank, world, device, dist):
    """BASELINE config 5: 1024-token int32 NdarrayCodec rows with predicate
    + shuffling queue, HBM cache."""
    from petastorm_amd import make_batch_reader
    from petastorm_amd.predicates import in_lambda
    from petastorm_amd.pytorch import BatchedDataLoader
    from petastorm_amd.test_util.dataset_gen import create_sequence_dataset

    # rowgroup sweep r2.7: 25k-row groups beat 6.25k by ~16%; 16 groups
    # keep 8-GPU shards equal (2 each)
    n_rows = args.rows or 400_000
    seq_rg = int(os.environ.get('PSA_SEQ_RG', '25000'))
    wpk = os.environ.get('PSA_WRITER_PAGE_KB', '256')
    url = _dataset_dir('seq_{}_g{}_w{}'.format(n_rows, seq_rg, wpk), rank,
                       dist,
                       lambda u: create_sequence_dataset(
                           u, num_rows=n_rows, rows_per_rowgroup=seq_rg))
    pred = in_lambda(['source'], lambda v: v['source'] != 3)  # keep 3/4
    reader = make_batch_reader(
        url, device=str(device), num_epochs=None, shuffle_row_groups=True,
        seed=5, predicate=pred,
        cur_shard=rank if world > 1 else None,
        shard_count=world if world > 1 else None,
        gpu_options=dict(cache_type='hbm', cache_size_limit=64 << 30,
                         decode_streams=int(os.environ.get(
                             'PSA_DECODE_STREAMS', '8'))))
    # batch sweep r2.8: 4096-row batches lift the warm tier 3.7x over
    # 1024 (loader slicing amortization); queue holds 4 batches
    loader = BatchedDataLoader(reader, batch_size=args.batch_size * 16,
                               shuffling_queue_capacity=args.batch_size * 64,
                               seed=3)
    it = iter(loader)

    def step():
        b = next(it)
        return b['tokens'].shape[0]

    # this config MEASURES the HBM-cache tier (BASELINE config 5: epoch
    # re-reads served from HBM): warmup must cover the cold first epoch so
    # the timed steps hit the cache steady state.  The raised warmup is
    # reported in the JSON line.
    per_rank_rows = n_rows // max(1, world)
    batch_rows = args.batch_size * 16
    args.warmup = max(args.warmup,
                      (per_rank_rows + batch_rows - 1) // batch_rows + 8)
    result = _run_timed(args, step, device, dist, world)
    reader.stop()
    reader.join()
    if os.environ.get('PSA_TIMING') == '1' and rank == 0:
        print('stage_times:', reader.diagnostics.get('stage_times'),
              file=sys.stderr)
    return result, {
        'model': 'SequenceSchema (1024-token int32 NdarrayCodec + predicate '
                 '+ shuffling queue, HBM cache)',
        'global_batch': args.batch_size * 16 * world,
        'seq_len': 1024,
        'parallelism': 'dp{}'.format(world),
        'pipeline': 'make_batch_reader(device=cuda) + HBM rowgroup cache',
    }


def _run_timed(args, step_fn, device, dist, world):
    # warmup
    warm_samples = 0
    for _ in range(args.warmup):
        warm_samples += step_fn()
    _sync(device)
    # Calibrate how many loader batches form one timed step so the timed
    # region spans >= --min-region seconds regardless of the requested step
    # count (a 20-step imagenet run is ~22 ms of GPU time otherwise —
    # invisible to SMI sampling and too short to trust).
    batches_per_step = 1
    if args.min_region > 0:
        t0 = time.perf_counter()
        cal = 0
        while cal < 3 or time.perf_counter() - t0 < 0.25:
            step_fn()
            cal += 1
            if cal >= 10000:
                break
        _sync(device)
        per_batch = (time.perf_counter() - t0) / cal
        # 1.5x headroom: steady-state batches run faster than the
        # calibration window (caches warm, pipeline full; measured up to
        # ~1.35x on the fused imagenet path)
        batches_per_step = max(
            1, int(np.ceil(1.5 * args.min_region /
                           (args.steps * per_batch))))
        if dist is not None:
            # all ranks must agree or lock-step collectives skew
            t = torch.tensor([batches_per_step], dtype=torch.int64)
            if torch.cuda.is_available():
                t = t.to(device)
            dist.all_reduce(t, op=dist.ReduceOp.MAX)
            batches_per_step = int(t.item())
    # The calibration probe can overestimate per-batch cost under transient
    # contention, which would undershoot the region; if that happens, scale
    # batches_per_step up and re-time (every rank sees the same MAX-reduced
    # elapsed, so they rescale in lock-step).
    attempts = 0
    while True:
        _barrier(dist)
        _sync(device)
        t0 = time.perf_counter()
        samples = 0
        for _ in range(args.steps):
            for _ in range(batches_per_step):
                samples += step_fn()
        _sync(device)
        _barrier(dist)
        elapsed = time.perf_counter() - t0
        # max elapsed over ranks (the slowest rank defines job time)
        if dist is not None:
            t = torch.tensor([elapsed], dtype=torch.float64)
            if torch.cuda.is_available():
                t = t.to(device)
            dist.all_reduce(t, op=dist.ReduceOp.MAX)
            elapsed = float(t.item())
            s = torch.tensor([samples], dtype=torch.float64)
            if torch.cuda.is_available():
                s = s.to(device)
            dist.all_reduce(s, op=dist.ReduceOp.SUM)
            samples = int(s.item())
        attempts += 1
        if args.min_region <= 0 or elapsed >= args.min_region \
                or attempts >= 3:
            break
        scale = max(2.0, 1.5 * args.min_region / max(elapsed, 1e-9))
        batches_per_step = int(np.ceil(batches_per_step * scale))
    return {'elapsed_s': elapsed, 'samples': samples,
            'ms_per_step': elapsed * 1000.0 / args.steps,
            'batches_per_step': batches_per_step}


CONFIGS = {
    'imagenet': bench_imagenet,
    'scalar': bench_scalar,
    'helloworld': bench_helloworld_cpu,
    'ngram': bench_ngram,
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--gpus', type=int, default=1)
    ap.add_argument('--steps', type=int, default=30)
    ap.add_argument('--warmup', type=int, default=10)
    ap.add_argument('--batch-size', type=int, default=256)
    ap.add_argument('--config', choices=sorted(CONFIGS), default='imagenet')
    ap.add_argument('--rows', type=int, default=None)
    ap.add_argument('--min-region', type=float,
                    default=float(os.environ.get('PSA_MIN_REGION', '5.0')),
                    help='minimum timed-region seconds: each of the K steps '
                         'consumes as many loader batches as needed to span '
                         'this (0 disables; one step == one batch then)')
    args = ap.parse_args()

    if args.gpus > 1 and 'WORLD_SIZE' not in os.environ:
        sys.exit(_self_spawn(args))

    si = os.environ.get('PSA_SWITCH_INTERVAL')
    if si:
        sys.setswitchinterval(float(si))
    rank, world, local = _dist_env()
    dist = _init_dist(world)
    if torch.cuda.is_available():
        # validation aid: multiple ranks can share one GPU (with a gloo
        # process group) on single-GPU test boxes
        local = min(local, torch.cuda.device_count() - 1)
        torch.cuda.set_device(local)
        device = torch.device('cuda', local)
    else:
        device = torch.device('cpu')
        if args.config in ('imagenet', 'scalar', 'ngram'):
            print('ERROR: config {!r} needs a GPU'.format(args.config),
                  file=sys.stderr)
            if args.config != 'helloworld':
                sys.exit(2)

    result, config = CONFIGS[args.config](args, rank, world, device, dist)

    if rank == 0:
        value = result['samples'] / result['elapsed_s']
        config = dict(config)
        bps = result.get('batches_per_step', 1)
        config['batches_per_step'] = bps
        config['global_batch'] = config.get('global_batch', 0) * bps
        out = {
            'metric': 'samples/sec/node (make_batch_reader->PyTorch '
                      'DataLoader), {} config'.format(args.config),
            'value': round(value, 2),
            'unit': 'samples/sec',
            'n_gpus': world,
            'steps': args.steps,
            'warmup': args.warmup,
            'ms_per_step': round(result['ms_per_step'], 3),
            'timed_region_s': round(result['elapsed_s'], 3),
            'higher_is_better': True,
            'scaling': 'weak',
            'vs_baseline': round(value / BASELINE_SAMPLES_PER_SEC, 3),
            'dtype': 'fp32',
            'data': 'synthetic (random-init {} rows, generated untimed)'
                    .format(args.rows or 'default'),
            'config': config,
        }
        print(json.dumps(out))
    if dist is not None:
        dist.destroy_process_group()


if __name__ == '__main__':
    main()
