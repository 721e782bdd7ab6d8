"""Row-group indexes + selectors (parity: reference tests/test_end_to_end.py
selector coverage + etl/rowgroup_indexing tests)."""
import pytest

from petastorm_amd import make_reader
from petastorm_amd.etl.rowgroup_indexing import (FieldNotNullIndexer,
                                                 SingleFieldIndexer,
                                                 build_rowgroup_index,
                                                 load_rowgroup_indexes)
from petastorm_amd.fs_utils import get_filesystem_and_path_or_paths
from petastorm_amd.selectors import (IntersectIndexSelector,
                                     SingleIndexSelector, UnionIndexSelector)


@pytest.fixture(scope='module')
def indexed_dataset(tmp_path_factory):
    from petastorm_amd.test_util.dataset_gen import create_test_dataset
    path = tmp_path_factory.mktemp('indexed_ds')
    url = 'file://' + str(path)
    rows = create_test_dataset(url, num_rows=60, num_files=2,
                               rowgroup_size_mb=0.02, seed=0)
    build_rowgroup_index(url, [
        SingleFieldIndexer('id2_index', 'id2'),
        SingleFieldIndexer('sensor_index', 'sensor_name'),
        FieldNotNullIndexer('nullable_index', 'matrix_nullable'),
    ])
    return {'url': url, 'rows': rows}


def test_index_persistence(indexed_dataset):
    fs, path = get_filesystem_and_path_or_paths(indexed_dataset['url'])
    indexes = load_rowgroup_indexes(fs, path)
    assert set(indexes) == {'id2_index', 'sensor_index', 'nullable_index'}
    assert indexes['id2_index'].get_row_group_indexes(0)


def test_single_index_selector(indexed_dataset):
    sel = SingleIndexSelector('sensor_index', ['sensor-1'])
    with make_reader(indexed_dataset['url'], reader_pool_type='dummy',
                     shuffle_row_groups=False, rowgroup_selector=sel) as r:
        rows = list(r)
    got_ids = {int(x.id) for x in rows}
    expected = {int(s['id']) for s in indexed_dataset['rows']
                if s['sensor_name'] == 'sensor-1'}
    # selector is row-GROUP level: all matching ids must be present
    assert expected <= got_ids


def test_intersect_and_union_selectors(indexed_dataset):
    s1 = SingleIndexSelector('id2_index', [1])
    s2 = SingleIndexSelector('sensor_index', ['sensor-0'])
    fs, path = get_filesystem_and_path_or_paths(indexed_dataset['url'])
    indexes = load_rowgroup_indexes(fs, path)
    inter = IntersectIndexSelector([s1, s2]).select_row_groups(indexes)
    union = UnionIndexSelector([s1, s2]).select_row_groups(indexes)
    assert inter <= union
    assert union == (s1.select_row_groups(indexes) |
                     s2.select_row_groups(indexes))


def test_not_null_indexer(indexed_dataset):
    fs, path = get_filesystem_and_path_or_paths(indexed_dataset['url'])
    indexes = load_rowgroup_indexes(fs, path)
    assert indexes['nullable_index'].get_row_group_indexes()


def test_missing_index_raises(indexed_dataset):
    sel = SingleIndexSelector('missing_index', [1])
    with pytest.raises(ValueError):
        make_reader(indexed_dataset['url'], reader_pool_type='dummy',
                    rowgroup_selector=sel)


def test_gpu_reader_accepts_rowgroup_selector(indexed_dataset):
    """GpuBatchReader applies index selectors at planning time (CPU-only
    check: the constructor's piece filtering needs no GPU)."""
    from petastorm_amd.gpu.reader import GpuBatchReader
    fs, path = get_filesystem_and_path_or_paths(indexed_dataset['url'])
    all_r = GpuBatchReader(fs, path, shuffle_row_groups=False,
                           device='cpu')
    sel_r = GpuBatchReader(fs, path, shuffle_row_groups=False, device='cpu',
                           rowgroup_selector=SingleIndexSelector(
                               'id2_index', [1]))
    assert 0 < len(sel_r._pieces) < len(all_r._pieces)


def test_gpu_route_rejects_unsupported_options(indexed_dataset):
    from petastorm_amd import make_batch_reader
    with pytest.raises(NotImplementedError, match='shuffle_row_drop'):
        make_batch_reader(indexed_dataset['url'], device='cuda',
                          shuffle_row_drop_partitions=2)
    with pytest.raises(NotImplementedError, match='cache_type'):
        make_batch_reader(indexed_dataset['url'], device='cuda',
                          cache_type='local-disk')
