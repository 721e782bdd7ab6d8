"""Sustained-throughput stability probe: long reader loops with RSS / HBM
tracking.  Run on a GPU box:

    python tools/stability_probe.py [--minutes 3]

Exercises the three GPU bench pipelines back to back and reports memory
growth between the first and last measurement windows — growth beyond the
noise threshold exits non-zero so the run can gate a release.
"""
import argparse
import os
import resource
import sys
import time

import torch


def _rss_mb():
    return resource.getrusage(resource.RUSAGE_SELF).ru_maxrss / 1024.0


def _hbm_mb():
    return torch.cuda.memory_allocated() / (1 << 20)


def run_config(name, seconds):
    sys.argv = ['bench.py']
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    import bench
    import argparse as ap
    args = ap.Namespace(gpus=1, steps=10, warmup=2, batch_size=256,
                        config=name, rows=None)
    device = torch.device('cuda')

    # build the loader exactly as bench.py does, then loop it ourselves
    base, _, variant = name.partition('-')
    if variant:  # e.g. scalar-zstd / scalar-lz4: page-codec variants
        os.environ['PSA_SCALAR_COMPRESSION'] = variant
    else:
        os.environ.pop('PSA_SCALAR_COMPRESSION', None)
    if base == 'imagenet':
        fn = bench.bench_imagenet
    elif base == 'scalar':
        fn = bench.bench_scalar
    else:
        fn = bench.bench_ngram

    # bench_* functions run a fixed number of steps; instead reuse their
    # dataset + reader wiring by monkeypatching the timing loop
    windows = []

    def fake_run_timed(a, step_fn, dev, dist, world):
        t_end = time.time() + seconds
        steps = 0
        rows = 0
        win_t0 = time.time()
        win_rows = 0
        while time.time() < t_end:
            r = step_fn()
            rows += r
            win_rows += r
            steps += 1
            if time.time() - win_t0 > 5.0:
                windows.append((win_rows / (time.time() - win_t0),
                                _rss_mb(), _hbm_mb()))
                win_t0 = time.time()
                win_rows = 0
        return {'elapsed_s': seconds, 'samples': rows,
                'ms_per_step': seconds * 1000.0 / max(1, steps)}

    orig = bench._run_timed
    bench._run_timed = fake_run_timed
    try:
        fn(args, 0, 1, device, None)
    finally:
        bench._run_timed = orig
    return windows


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--minutes', type=float, default=3.0)
    p.add_argument('--configs', default='imagenet,scalar,scalar-zstd,'
                   'scalar-lz4,ngram',
                   help='comma list; scalar-<codec> runs the scalar '
                        'pipeline with that Parquet page compression')
    args = p.parse_args()
    configs = [c for c in args.configs.split(',') if c]
    per_cfg = args.minutes * 60 / len(configs)
    bad = 0
    for cfg in configs:
        w = run_config(cfg, per_cfg)
        if len(w) < 2:
            print(f'{cfg}: too few windows ({len(w)})')
            continue
        first, last = w[0], w[-1]
        thr = [x[0] for x in w]
        print(f'{cfg}: windows={len(w)} '
              f'throughput first={first[0]:,.0f}/s last={last[0]:,.0f}/s '
              f'min={min(thr):,.0f} max={max(thr):,.0f} | '
              f'RSS {first[1]:.0f}->{last[1]:.0f} MB | '
              f'HBM {first[2]:.0f}->{last[2]:.0f} MB')
        # ru_maxrss is a high-water mark; flag only large late growth
        if last[1] - first[1] > 500:
            print(f'{cfg}: RSS grew {last[1]-first[1]:.0f} MB'); bad += 1
        if last[2] - first[2] > 2048:
            print(f'{cfg}: HBM grew {last[2]-first[2]:.0f} MB'); bad += 1
        if last[0] < 0.5 * max(thr):
            print(f'{cfg}: throughput decayed to {last[0]:,.0f}'); bad += 1
    print('STABILITY', 'FAIL' if bad else 'PASS')
    sys.exit(1 if bad else 0)


if __name__ == '__main__':
    main()
