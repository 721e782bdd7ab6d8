"""Hive-partitioned stores: partition column materialization + predicate
pushdown (parity: reference tests/test_parquet_reader.py partitioned
coverage and reader.py:533-571 partition predicate)."""
import numpy as np
import pytest

from petastorm_amd import make_batch_reader, make_reader
from petastorm_amd.predicates import in_lambda, in_set


@pytest.fixture(scope='module')
def partitioned_dataset(tmp_path_factory):
    import pyarrow as pa
    import pyarrow.parquet as pq
    root = tmp_path_factory.mktemp('part_ds')
    rows_per_part = 50
    idx = 0
    for color in ('red', 'green', 'blue'):
        d = root / 'color={}'.format(color)
        d.mkdir()
        table = pa.table({
            'id': np.arange(idx, idx + rows_per_part, dtype=np.int64),
            'x': np.random.RandomState(idx).rand(rows_per_part),
        })
        pq.write_table(table, str(d / 'part-0.parquet'),
                       use_dictionary=False, row_group_size=25)
        idx += rows_per_part
    return {'url': 'file://' + str(root)}


def test_partition_column_materialized_batch(partitioned_dataset):
    with make_batch_reader(partitioned_dataset['url'],
                           reader_pool_type='dummy',
                           shuffle_row_groups=False) as r:
        assert 'color' in r.schema.fields
        batches = list(r)
    total = sum(len(b.id) for b in batches)
    assert total == 150
    colors = {c for b in batches for c in np.asarray(b.color).tolist()}
    assert colors == {'red', 'green', 'blue'}


def test_partition_column_materialized_rows(partitioned_dataset):
    with make_reader(partitioned_dataset['url'], reader_pool_type='dummy',
                     shuffle_row_groups=False) as r:
        rows = list(r)
    assert len(rows) == 150
    assert {row.color for row in rows} == {'red', 'green', 'blue'}


def test_partition_predicate_pushdown(partitioned_dataset):
    pred = in_set({'red'}, 'color')
    with make_reader(partitioned_dataset['url'], reader_pool_type='dummy',
                     shuffle_row_groups=False, predicate=pred) as r:
        rows = list(r)
        # pushdown: only red's row groups were ventilated (2 of 6)
        assert r.diagnostics['items_ventilated'] == 2
    assert len(rows) == 50
    assert all(row.color == 'red' for row in rows)
    assert all(int(row.id) < 50 for row in rows)


def test_partition_predicate_in_worker_for_batch(partitioned_dataset):
    pred = in_lambda(['color'], lambda v: v['color'] == 'blue')
    with make_reader(partitioned_dataset['url'], reader_pool_type='dummy',
                     shuffle_row_groups=False, predicate=pred) as r:
        rows = list(r)
    assert len(rows) == 50 and all(r_.color == 'blue' for r_ in rows)


def test_mixed_file_and_partition_predicate(tmp_path):
    """A predicate over one parquet column AND one hive-partition key works
    on both routes (reference piece.read(partitions=...) supports this,
    arrow_reader_worker.py:358)."""
    import os
    import numpy as np
    import pyarrow as pa
    import pyarrow.parquet as pq
    from petastorm_amd import make_batch_reader, make_reader
    from petastorm_amd.predicates import in_lambda

    d = str(tmp_path / 'mixed')
    for color in ('red', 'blue'):
        sub = os.path.join(d, 'color=%s' % color)
        os.makedirs(sub)
        t = pa.table({'id': pa.array(np.arange(100, dtype=np.int64))})
        pq.write_table(t, os.path.join(sub, 'f.parquet'), row_group_size=50)
    url = 'file://' + d

    pred_vec = in_lambda(['id', 'color'],
                         lambda v: (np.asarray(v['id']) % 2 == 0) &
                                   (np.asarray(v['color']) == 'red'))
    with make_batch_reader(url, predicate=pred_vec,
                           shuffle_row_groups=False) as r:
        rows = [(int(i), str(c)) for b in r
                for i, c in zip(b.id, b.color)]
    assert len(rows) == 50
    assert all(i % 2 == 0 and c == 'red' for i, c in rows)

    pred_row = in_lambda(['id', 'color'],
                         lambda v: v['id'] % 2 == 0 and v['color'] == 'red')
    with make_reader(url, predicate=pred_row,
                     shuffle_row_groups=False) as r:
        got = [(int(row.id), str(row.color)) for row in r]
    assert sorted(got) == sorted(rows)
