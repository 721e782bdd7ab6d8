"""URL resolution (parity: reference tests/test_fs_utils.py)."""
import pytest

from petastorm_amd.fs_utils import (get_filesystem_and_path_or_paths,
                                    normalize_dataset_url_or_urls,
                                    normalize_dir_url)


def test_normalize_dir_url():
    assert normalize_dir_url('file:///a/b/') == 'file:///a/b'
    with pytest.raises(ValueError):
        normalize_dir_url(123)


def test_normalize_url_list():
    assert normalize_dataset_url_or_urls(['file:///a/', 'file:///b']) == \
        ['file:///a', 'file:///b']
    with pytest.raises(ValueError):
        normalize_dataset_url_or_urls([])


def test_file_url_resolution(tmp_path):
    fs, path = get_filesystem_and_path_or_paths('file://' + str(tmp_path))
    assert path == str(tmp_path)
    assert fs.exists(str(tmp_path))


def test_bare_path_resolution(tmp_path):
    fs, path = get_filesystem_and_path_or_paths(str(tmp_path))
    assert path == str(tmp_path)


def test_mixed_schemes_raise():
    with pytest.raises(ValueError):
        get_filesystem_and_path_or_paths(['file:///a', 's3://bucket/b'])


def test_url_list_resolution(tmp_path):
    fs, paths = get_filesystem_and_path_or_paths(
        ['file://' + str(tmp_path), 'file://' + str(tmp_path)])
    assert paths == [str(tmp_path)] * 2


# ---------------------------------------------------------------------------
# remote-scheme + retry behavior
# ---------------------------------------------------------------------------
import time

import pytest

from petastorm_amd.fs_utils import RetryingFilesystem


class _FlakyFs(object):
    """Fails each wrapped call a configured number of times, then succeeds
    (models transient namenode/S3 failures; reference mocks HDFS the same
    way, hdfs/tests/test_hdfs_namenode.py:265-307)."""

    def __init__(self, failures):
        self.failures = failures
        self.calls = 0

    def exists(self, path):
        self.calls += 1
        if self.calls <= self.failures:
            raise IOError('transient failure #%d' % self.calls)
        return True

    def ls(self, path):
        self.calls += 1
        if self.calls <= self.failures:
            raise IOError('transient')
        return ['a', 'b']

    def size(self, path):  # not in the retryable allow-list
        self.calls += 1
        raise IOError('always fails')


def test_retrying_filesystem_recovers_from_transient_failures():
    fs = RetryingFilesystem(_FlakyFs(failures=2), attempts=3,
                            backoff_s=0.001)
    assert fs.exists('/x') is True


def test_retrying_filesystem_gives_up_after_attempts():
    flaky = _FlakyFs(failures=10)
    fs = RetryingFilesystem(flaky, attempts=3, backoff_s=0.001)
    with pytest.raises(IOError):
        fs.exists('/x')
    assert flaky.calls == 3


def test_retrying_filesystem_backoff_is_bounded():
    flaky = _FlakyFs(failures=10)
    fs = RetryingFilesystem(flaky, attempts=4, backoff_s=0.01)
    t0 = time.time()
    with pytest.raises(IOError):
        fs.ls('/x')
    assert time.time() - t0 < 2.0


def test_retrying_filesystem_passthrough_non_retryable():
    flaky = _FlakyFs(failures=0)
    fs = RetryingFilesystem(flaky, attempts=3, backoff_s=0.001)
    with pytest.raises(IOError):
        fs.size('/x')  # not retried
    assert flaky.calls == 1


def test_memory_scheme_end_to_end():
    """A non-file fsspec scheme through the whole write->read stack
    (memory:// — the only remote-like fs available offline)."""
    import numpy as np
    from petastorm_amd import make_batch_reader
    from petastorm_amd.fs_utils import get_filesystem_and_path_or_paths
    fs, path = get_filesystem_and_path_or_paths('memory://psa_test_ds')
    assert type(fs).__name__ == 'RetryingFilesystem'
    import pyarrow as pa
    import pyarrow.parquet as pq
    table = pa.table({'id': np.arange(50, dtype=np.int64),
                      'v': np.linspace(0, 1, 50)})
    fs.makedirs('/psa_test_ds', exist_ok=True)
    with fs.open('/psa_test_ds/p.parquet', 'wb') as f:
        pq.write_table(table, f, row_group_size=10)
    with make_batch_reader('memory://psa_test_ds', num_epochs=1,
                           shuffle_row_groups=False) as r:
        ids = np.concatenate([np.asarray(b.id) for b in r])
    assert sorted(ids.tolist()) == list(range(50))


def test_s3_netloc_path_quirk():
    """s3:// URLs keep the bucket in the path handed to fsspec
    (reference get_dataset_path, fs_utils.py:28-38)."""
    import fsspec
    from petastorm_amd.fs_utils import get_filesystem_and_path_or_paths
    try:
        fsspec.get_filesystem_class('s3')
    except (ImportError, ValueError):
        pytest.skip('s3fs not installed')
    fs, path = get_filesystem_and_path_or_paths('s3://bucket/key/dir')
    assert path == 'bucket/key/dir'
