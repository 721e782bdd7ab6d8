"""make_batch_reader over plain (non-petastorm) Parquet stores.

Parity: reference tests/test_parquet_reader.py:71-627.
"""
import numpy as np
import pytest

from petastorm_amd import make_batch_reader
from petastorm_amd.predicates import in_lambda


def _collect(batches):
    out = {}
    for b in batches:
        for name in b._fields:
            out.setdefault(name, []).append(getattr(b, name))
    return {k: np.concatenate(v) for k, v in out.items()}


def test_scalar_store_roundtrip(scalar_dataset):
    with make_batch_reader(scalar_dataset['url'], reader_pool_type='thread',
                           workers_count=3, shuffle_row_groups=False) as r:
        cols = _collect(list(r))
    src = scalar_dataset['cols']
    order = np.argsort(cols['id'])
    np.testing.assert_array_equal(cols['id'][order], src['id'])
    np.testing.assert_array_almost_equal(cols['f0'][order], src['f0'])
    np.testing.assert_array_equal(cols['i3'][order], src['i3'])
    assert cols['name'][order][0] == 'row-0'


def test_column_subset(scalar_dataset):
    with make_batch_reader(scalar_dataset['url'], reader_pool_type='dummy',
                           schema_fields=['id', 'f1'],
                           shuffle_row_groups=False) as r:
        b = next(iter(r))
    assert set(b._fields) == {'id', 'f1'}


def test_invalid_column_raises(scalar_dataset):
    with pytest.raises(ValueError):
        make_batch_reader(scalar_dataset['url'], reader_pool_type='dummy',
                          schema_fields=['nope_.*_col'])


def test_vectorized_predicate(scalar_dataset):
    pred = in_lambda(['id'], lambda v: v['id'] % 2 == 0)
    with make_batch_reader(scalar_dataset['url'], reader_pool_type='dummy',
                           predicate=pred, shuffle_row_groups=False) as r:
        cols = _collect(list(r))
    assert (cols['id'] % 2 == 0).all()
    assert len(cols['id']) == 250


def test_num_epochs_batches(scalar_dataset):
    with make_batch_reader(scalar_dataset['url'], reader_pool_type='thread',
                           num_epochs=2, shuffle_row_groups=False) as r:
        cols = _collect(list(r))
    assert len(cols['id']) == 1000


def test_seeded_shuffle_batches_deterministic(scalar_dataset):
    def ids(seed):
        with make_batch_reader(scalar_dataset['url'],
                               reader_pool_type='thread', workers_count=2,
                               shuffle_row_groups=True, seed=seed) as r:
            return _collect(list(r))['id'].tolist()

    assert ids(11) == ids(11)
    assert ids(11) != ids(12)


def test_shuffle_rows_within_rowgroup(scalar_dataset):
    with make_batch_reader(scalar_dataset['url'], reader_pool_type='dummy',
                           shuffle_row_groups=False, shuffle_rows=True,
                           seed=3) as r:
        b = next(iter(r))
    assert b.id.tolist() != sorted(b.id.tolist())
    assert sorted(b.id.tolist()) == list(range(len(b.id)))


def test_cache_and_shuffle_across_epochs(scalar_dataset, tmp_path):
    """Cached row-groups re-shuffle correctly across epochs (reference
    tests/test_parquet_reader.py shuffle+cache interaction)."""
    kwargs = dict(reader_pool_type='thread', workers_count=2,
                  shuffle_row_groups=True, seed=4, num_epochs=2,
                  cache_type='local-disk',
                  cache_location=str(tmp_path / 'cache'),
                  cache_size_limit=200 << 20, cache_row_size_estimate=1024,
                  schema_fields=['id'])
    with make_batch_reader(scalar_dataset['url'], **kwargs) as r:
        ids = _collect(list(r))['id']
    assert len(ids) == 1000
    np.testing.assert_array_equal(np.sort(ids), np.repeat(np.arange(500), 2))
    # second reader: cache hits must yield identical content
    with make_batch_reader(scalar_dataset['url'], **kwargs) as r:
        ids2 = _collect(list(r))['id']
    np.testing.assert_array_equal(ids, ids2)


def test_zero_rows_after_filter_cached(scalar_dataset, tmp_path):
    from petastorm_amd.predicates import in_lambda
    pred = in_lambda(['id'], lambda v: v['id'] < 0)  # matches nothing
    kwargs = dict(reader_pool_type='dummy', shuffle_row_groups=False,
                  predicate=pred)
    with make_batch_reader(scalar_dataset['url'], **kwargs) as r:
        assert list(r) == []


def test_filters_statistics_pruning(scalar_dataset):
    """pyarrow-style `filters`: row groups pruned by min/max statistics
    (reference forwards filters to pq.ParquetDataset, reader.py:431-433)."""
    with make_batch_reader(scalar_dataset['url'], reader_pool_type='dummy',
                           shuffle_row_groups=False,
                           schema_fields=['id'],
                           filters=[('id', '>=', 400)]) as r:
        ids = _collect(list(r))['id']
        # only the last row group (ids 400..499) should be ventilated
        assert r.diagnostics['items_ventilated'] == 1
    assert set(ids) == set(range(400, 500))


def test_filters_dnf_or(scalar_dataset):
    dnf = [[('id', '<', 100)], [('id', '>=', 400)]]
    with make_batch_reader(scalar_dataset['url'], reader_pool_type='dummy',
                           shuffle_row_groups=False, schema_fields=['id'],
                           filters=dnf) as r:
        ids = _collect(list(r))['id']
        assert r.diagnostics['items_ventilated'] == 2
    assert set(ids) == set(range(0, 100)) | set(range(400, 500))


# ---------------------------------------------------------------------------
# full scalar-type surface + many-columns stores (reference
# tests/test_common.py:161-294, tests/test_parquet_reader.py)
# ---------------------------------------------------------------------------

def test_rich_scalar_store_types(tmp_path):
    """date/timestamp/strings/float64/fixed-size list/nested struct all
    come back with the reference's observable type mapping."""
    import datetime
    from petastorm_amd import make_batch_reader
    from petastorm_amd.test_util.dataset_gen import create_rich_scalar_dataset
    url = 'file://' + str(tmp_path / 'rich')
    rows = create_rich_scalar_dataset(url, num_rows=60, rowgroup_size=20)
    got = {}
    with make_batch_reader(url, shuffle_row_groups=False) as r:
        for b in r:
            ids = np.asarray(b.id)
            for i, rid in enumerate(ids):
                got[int(rid)] = {f: np.asarray(getattr(b, f))[i]
                                 for f in b._fields}
    assert len(got) == 60
    for src in rows:
        row = got[int(src['id'])]
        assert row['string'] == src['string']
        assert row['float64'] == src['float64']
        np.testing.assert_array_equal(row['int_fixed_size_list'],
                                      src['int_fixed_size_list'])
        # datetime.date -> datetime64[D]-compatible value
        assert np.datetime64(row['datetime'], 'D') == \
            np.datetime64(src['datetime'], 'D')
        assert np.datetime64(row['timestamp'], 'us') == \
            np.datetime64(src['timestamp'], 'us')
        # nested structs are omitted from the inferred schema with a
        # warning — the reference does the same (unischema.py:303
        # omit_unsupported_fields=True by default)
        assert 'nested_struct' not in row


def test_many_columns_store(tmp_path):
    """1000 int32 columns (reference many_columns_non_petastorm_dataset):
    full read and a 3-column subset."""
    from petastorm_amd import make_batch_reader
    from petastorm_amd.test_util.dataset_gen import create_many_columns_dataset
    url = 'file://' + str(tmp_path / 'wide')
    cols = create_many_columns_dataset(url, num_rows=25, num_columns=1000)
    with make_batch_reader(url, shuffle_row_groups=False) as r:
        b = next(iter(r))
        assert len(b._fields) == 1000
        np.testing.assert_array_equal(np.sort(np.asarray(b.col_0)),
                                      cols['col_0'][:len(b.col_0)])
    with make_batch_reader(url, shuffle_row_groups=False,
                           schema_fields=['col_1', 'col_42',
                                          'col_999']) as r:
        b = next(iter(r))
        assert sorted(b._fields) == ['col_1', 'col_42', 'col_999']


def test_weighted_sampling_over_batch_readers(scalar_dataset, tmp_path):
    """WeightedSamplingReader mixes batched readers too (reference
    weighted_sampling_reader.py checks batched_output compatibility)."""
    from petastorm_amd import make_batch_reader
    from petastorm_amd.weighted_sampling_reader import WeightedSamplingReader
    from petastorm_amd.test_util.dataset_gen import create_scalar_dataset
    url2 = 'file://' + str(tmp_path / 'mix2')
    create_scalar_dataset(url2, num_rows=400, rowgroup_size=100, seed=9)
    r1 = make_batch_reader(scalar_dataset['url'], num_epochs=None,
                           schema_fields=['id'], shuffle_row_groups=False)
    r2 = make_batch_reader(url2, num_epochs=None, schema_fields=['id'],
                           shuffle_row_groups=False)
    mixed = WeightedSamplingReader([r1, r2], [0.5, 0.5])
    batches = [next(mixed) for _ in range(20)]
    assert all(hasattr(b, 'id') for b in batches)
    r1.stop(); r1.join(); r2.stop(); r2.join()


def test_upstream_compat_kwargs(scalar_dataset):
    """Upstream call sites pass shard_seed/hdfs_driver/zmq_copy_buffers/
    pyarrow_serialize/filesystem — accepted (deprecation warnings for
    no-effect knobs; filesystem is honored)."""
    import warnings
    import fsspec
    from petastorm_amd import make_batch_reader
    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter('always')
        with make_batch_reader(scalar_dataset['url'], shard_seed=7,
                               hdfs_driver='libhdfs3',
                               zmq_copy_buffers=True,
                               shuffle_row_groups=True) as r:
            ids = _collect(list(r))['id']
        dep = [x for x in w if issubclass(x.category, DeprecationWarning)]
    assert len(ids) == 500
    assert len(dep) == 3  # shard_seed + hdfs_driver + zmq_copy_buffers
    fs = fsspec.filesystem('file')
    with make_batch_reader(scalar_dataset['url'], filesystem=fs,
                           shuffle_row_groups=False) as r:
        assert len(_collect(list(r))['id']) == 500


def test_weighted_sampling_seeded_determinism():
    """Same seed -> same source-reader draw sequence (seeding is an
    extension over the reference; the compat checks still apply)."""
    import numpy as np
    from petastorm_amd.unischema import Unischema, UnischemaField
    from petastorm_amd.weighted_sampling_reader import WeightedSamplingReader

    schema = Unischema('S', [UnischemaField('id', np.int64, (), None,
                                            False)])

    class _Tagged(object):
        batched_output = False
        ngram = None
        last_row_consumed = False

        def __init__(self, tag):
            self.tag = tag
            self.schema = schema

        def __next__(self):
            return self.tag

        def stop(self):
            pass

        def join(self):
            pass

    def run(seed):
        mixed = WeightedSamplingReader([_Tagged('a'), _Tagged('b')],
                                       [0.7, 0.3], seed=seed)
        return [next(mixed) for _ in range(40)]

    assert run(5) == run(5)
    assert run(5) != run(6)
    assert set(run(5)) == {'a', 'b'}


def test_seeded_thread_pool_order_matches_dummy(scalar_dataset):
    """With a seed, the thread pool's round-robin readout reproduces the
    dummy pool's (inline) batch order exactly — the determinism guarantee
    state_dict's fast-forward relies on (reference thread_pool.py:181-199
    round-robin rationale)."""
    def ids(pool):
        from petastorm_amd import make_batch_reader
        with make_batch_reader(scalar_dataset['url'], reader_pool_type=pool,
                               workers_count=3, shuffle_row_groups=True,
                               seed=31, num_epochs=2) as r:
            return [int(b.id[0]) for b in r]  # first id of each batch

    assert ids('thread') == ids('dummy')
