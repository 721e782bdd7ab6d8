"""Randomized reader-configuration soak.

Samples random combinations of reader options (pool, shuffle flags, seed,
epochs, predicate, sharding, column subsets, batch/row route) against a
generated store and verifies exact row coverage for each.  A
configuration-space complement to the per-feature tests.

    python tools/config_soak.py [--configs N] [--seed S]
"""
import argparse
import os
import sys
import tempfile

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..'))


def run_one(url, rng, n_rows):
    from petastorm_amd import make_batch_reader, make_reader
    from petastorm_amd.predicates import in_lambda

    route = rng.choice(['row', 'batch'])
    pool = rng.choice(['dummy', 'thread'])
    shuffle_rg = bool(rng.randint(2))
    shuffle_rows = bool(rng.randint(2))
    seed = int(rng.randint(10000))
    epochs = int(rng.choice([1, 2]))
    use_pred = bool(rng.randint(2))
    use_shard = bool(rng.randint(2))
    subset = bool(rng.randint(2))
    kw = dict(shuffle_row_groups=shuffle_rg, seed=seed, num_epochs=epochs,
              shuffle_rows=shuffle_rows)
    if subset:
        kw['schema_fields'] = ['id', 'f0']
    expected = set(range(n_rows))
    if use_pred:
        mod = int(rng.choice([2, 3]))
        if route == 'batch':
            kw['predicate'] = in_lambda(
                ['id'], lambda v, m=mod: np.asarray(v['id']) % m == 0)
        else:
            kw['predicate'] = in_lambda(
                ['id'], lambda v, m=mod: v['id'] % m == 0)
        expected = {i for i in expected if i % mod == 0}
    shard_count = None
    if use_shard:
        shard_count = int(rng.choice([2, 4]))
        kw['cur_shard'] = int(rng.randint(shard_count))
        kw['shard_count'] = shard_count
    desc = ('route={} pool={} rg={} rows={} seed={} ep={} pred={} '
            'shard={}/{} subset={}'.format(
                route, pool, shuffle_rg, shuffle_rows, seed, epochs,
                use_pred, kw.get('cur_shard'), shard_count, subset))
    ids = []
    if route == 'batch':
        with make_batch_reader(url, **kw) as r:
            for b in r:
                ids.extend(int(x) for x in b.id)
    else:
        kw['reader_pool_type'] = pool
        with make_reader(url, **kw) as r:
            ids.extend(int(row.id) for row in r)
    got = set(ids)
    # sharding partitions row GROUPS: each id appears in exactly one shard,
    # so got must be a subset; without sharding, exact coverage per epoch
    if shard_count is None:
        assert got == expected, (desc, len(got), len(expected))
        assert len(ids) == epochs * len(expected), (desc, len(ids))
    else:
        assert got <= expected, desc
        assert len(ids) == epochs * len(got), (desc, len(ids))
    return desc


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--configs', type=int, default=40)
    ap.add_argument('--seed', type=int, default=0)
    args = ap.parse_args()
    from petastorm_amd.test_util.dataset_gen import create_scalar_dataset
    url = 'file://' + tempfile.mkdtemp(prefix='soak_')
    n_rows = 600
    create_scalar_dataset(url, num_rows=n_rows, rowgroup_size=75)  # 8 rgs
    rng = np.random.RandomState(args.seed)
    for i in range(args.configs):
        desc = run_one(url, rng, n_rows)
        print('[{:3d}] OK {}'.format(i, desc), flush=True)
    print('ALL OK ({} configs)'.format(args.configs))


if __name__ == '__main__':
    main()
