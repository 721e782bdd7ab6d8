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


# ---------------------------------------------------------------------------
# HbmCache eviction logic (CPU-tensor unit tests; GPU usage covered by
# tests/test_gpu_decode.py::test_gpu_batch_reader_hbm_cache_and_epochs)
# ---------------------------------------------------------------------------

def test_hbm_cache_lru_eviction_and_budget():
    import torch
    from petastorm_amd.gpu.hbm_cache import HbmCache
    one_mb = torch.zeros(1 << 20, dtype=torch.uint8)
    cache = HbmCache(size_limit_bytes=3 << 20)
    for k in 'abc':
        cache.get(k, lambda: {'x': one_mb.clone()})
    assert cache.size_bytes == 3 << 20
    cache.get('a', lambda: (_ for _ in ()).throw(AssertionError('hit!')))
    # inserting d evicts the LRU entry: 'b' (a was refreshed)
    cache.get('d', lambda: {'x': one_mb.clone()})
    assert 'b' not in cache._store and 'a' in cache._store
    assert cache.size_bytes == 3 << 20
    # value over budget: served but never cached
    big = {'x': torch.zeros(4 << 20, dtype=torch.uint8)}
    cache.get('huge', lambda: big)
    assert 'huge' not in cache._store
    assert cache.hits == 1 and cache.misses == 5
    cache.cleanup()
    assert cache.size_bytes == 0


def test_corrupt_cache_entry_is_a_miss(tmp_path):
    """A truncated/garbage entry must refill, not raise (partial writes
    happen on crash)."""
    from petastorm_amd.cache import LocalDiskCache
    c = LocalDiskCache(str(tmp_path / 'c'), 10 << 20)
    assert c.get('k', lambda: [1, 2, 3]) == [1, 2, 3]
    # find the entry file and stomp it
    import glob
    (entry,) = glob.glob(str(tmp_path / 'c' / 'shard-*' / '*.pkl'))
    for garbage in (b'', b'\x80', b'\x80\x04garbage', b'not a pickle'):
        with open(entry, 'wb') as f:
            f.write(garbage)
        assert c.get('k', lambda: ['refilled']) == ['refilled']
