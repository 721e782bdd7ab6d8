"""LocalDiskCache (parity: reference tests/test_disk_cache.py:37-194)."""
import numpy as np
import pytest

from petastorm_amd.cache import LocalDiskCache, NullCache


def test_null_cache_always_fills():
    c = NullCache()
    calls = []
    assert c.get('k', lambda: calls.append(1) or 42) == 42
    assert c.get('k', lambda: calls.append(1) or 43) == 43
    assert len(calls) == 2


def test_disk_cache_hit(tmp_path):
    c = LocalDiskCache(str(tmp_path / 'c'), 10 << 20)
    calls = []

    def fill():
        calls.append(1)
        return {'a': np.arange(10)}

    v1 = c.get('key1', fill)
    v2 = c.get('key1', fill)
    assert len(calls) == 1
    np.testing.assert_array_equal(v1['a'], v2['a'])


def test_disk_cache_eviction(tmp_path):
    c = LocalDiskCache(str(tmp_path / 'c'), 600 * 1024, shards=2)
    for i in range(40):
        c.get('key{}'.format(i), lambda i=i: np.zeros(8192, dtype=np.uint8))
    assert c.size_bytes() <= 600 * 1024 + 2 * 8192


def test_disk_cache_capacity_sanity(tmp_path):
    with pytest.raises(ValueError):
        LocalDiskCache(str(tmp_path / 'c'), 1024, expected_row_size_bytes=10000)


def test_disk_cache_cleanup(tmp_path):
    import os
    path = str(tmp_path / 'c')
    c = LocalDiskCache(path, 1 << 20, cleanup=True)
    c.get('k', lambda: 1)
    assert os.path.exists(path)
    c.cleanup()
    assert not os.path.exists(path)
