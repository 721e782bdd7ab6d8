

def test_cpu_reader_state_dict_resume(tmp_path, scalar_dataset):
    """Deterministic fast-forward resume on the CPU pool path (the GPU
    reader has an exact-cursor state_dict; reference has neither)."""
    from petastorm_amd import make_reader
    url = scalar_dataset['url']
    kwargs = dict(reader_pool_type='thread', workers_count=3,
                  shuffle_row_groups=True, seed=11, num_epochs=2)
    with make_reader(url, **kwargs) as r1:
        first = [int(next(r1).id) for _ in range(120)]
        state = r1.state_dict()
        rest1 = [int(row.id) for row in r1]
    with make_reader(url, **kwargs) as r2:
        r2.load_state_dict(state)
        rest2 = [int(row.id) for row in r2]
    assert state['rows_consumed'] == 120
    assert rest1 == rest2
    assert len(first) + len(rest1) == 2 * 500


def test_cpu_reader_state_dict_requires_determinism(scalar_dataset):
    from petastorm_amd import make_reader
    import pytest as _pytest
    with make_reader(scalar_dataset['url'], shuffle_row_groups=True,
                     seed=None) as r:
        with _pytest.raises(NotImplementedError):
            r.state_dict()


class _CountingCache:
    """Records which row-group keys were decoded (fill invoked)."""

    def __init__(self):
        self.filled = []

    def get(self, key, fill):
        self.filled.append(key)
        return fill()

    def cleanup(self):
        pass


def test_state_dict_fast_cursor_skips_whole_rowgroups(tmp_path):
    """Plain config restore uses the O(1) item cursor: row groups before
    the checkpoint are never decoded by the fresh reader."""
    import numpy as np
    from petastorm_amd import make_reader
    from petastorm_amd.reader import Reader
    from petastorm_amd.workers.row_worker import RowReaderWorker
    from petastorm_amd.workers_pool.thread_pool import ThreadPool
    from petastorm_amd.fs_utils import get_filesystem_and_path_or_paths
    from petastorm_amd.test_util.dataset_gen import create_scalar_dataset

    url = 'file://' + str(tmp_path / 'fastskip')
    create_scalar_dataset(url, num_rows=1000, rowgroup_size=100)  # 10 rgs

    def build(cache=None):
        fs, path = get_filesystem_and_path_or_paths(url)
        return Reader(fs, path, worker_class=RowReaderWorker,
                      reader_pool=ThreadPool(3), schema_fields=['id'],
                      shuffle_row_groups=True, seed=77, num_epochs=2,
                      cache=cache)

    with build() as r1:
        first = [int(next(r1).id) for _ in range(437)]
        state = r1.state_dict()
        rest1 = [int(row.id) for row in r1]

    cache = _CountingCache()
    with build(cache=cache) as r2:
        r2.load_state_dict(state)
        rest2 = [int(row.id) for row in r2]
    assert rest2 == rest1
    # 437 rows = 4 whole skipped groups + 37 rows into the 5th: across the
    # ENTIRE restored run only the remaining groups decode (6 of epoch 1 +
    # 10 of epoch 2); a replay restore would decode all 20
    assert len(cache.filled) <= 16, (len(cache.filled), cache.filled)


def test_state_dict_fast_cursor_batch_mode(scalar_dataset):
    from petastorm_amd import make_batch_reader
    with make_batch_reader(scalar_dataset['url'], shuffle_row_groups=True,
                           seed=5, num_epochs=2) as r1:
        for _ in range(3):
            next(r1)
        state = r1.state_dict()
        rest1 = [b.id.sum() for b in r1]
    with make_batch_reader(scalar_dataset['url'], shuffle_row_groups=True,
                           seed=5, num_epochs=2) as r2:
        r2.load_state_dict(state)
        rest2 = [b.id.sum() for b in r2]
    assert [float(x) for x in rest1] == [float(x) for x in rest2]

def test_state_dict_with_row_filtering_transform(scalar_dataset):
    """A TransformSpec func that drops rows invalidates the metadata row
    counts, so restore must take the replay path and still be exact."""
    import numpy as np
    from petastorm_amd import make_reader
    from petastorm_amd.transform import TransformSpec

    def keep_even(row):
        return row if int(row['id']) % 2 == 0 else None

    ts = TransformSpec(func=keep_even)
    kwargs = dict(reader_pool_type='dummy', shuffle_row_groups=True,
                  seed=21, num_epochs=2, transform_spec=ts)
    with make_reader(scalar_dataset['url'], **kwargs) as r1:
        assert not r1._fast_skip_ok
        first = [int(next(r1).id) for _ in range(80)]
        state = r1.state_dict()
        rest1 = [int(row.id) for row in r1]
    with make_reader(scalar_dataset['url'], **kwargs) as r2:
        r2.load_state_dict(state)
        rest2 = [int(row.id) for row in r2]
    assert all(v % 2 == 0 for v in first + rest1)
    assert rest1 == rest2
