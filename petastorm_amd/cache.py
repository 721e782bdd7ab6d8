"""Read-through row-group caches.

Parity: /root/reference/petastorm/cache.py:21-39 (CacheBase/NullCache) and
/root/reference/petastorm/local_disk_cache.py:24-83 (LocalDiskCache over
``diskcache.FanoutCache``).

This framework implements its own sharded on-disk cache (no diskcache
dependency): one file per entry under ``shard-XX/`` directories, pickled
values, least-recently-stored eviction driven by a per-shard size budget —
the same observable contract (get-or-fill, size ceiling, cleanup()).

An HBM-resident cache tier for decoded row-groups on the GPU path lives in
``petastorm_amd.gpu.hbm_cache`` (reference has no GPU tier; its equivalent
role is local_disk_cache staging decoded rowgroups, which on MI355X belongs
in the 288 GB HBM3E).
"""

import hashlib
import os
import pickle
import shutil
import threading


class CacheBase(object):
    def get(self, key, fill_cache_func):
        raise NotImplementedError()

    def cleanup(self):
        pass


class NullCache(CacheBase):
    """Pass-through cache (reference cache.py:30-39)."""

    def get(self, key, fill_cache_func):
        return fill_cache_func()


class LocalDiskCache(CacheBase):
    """Sharded on-disk KV cache of decoded row-groups.

    :param path: cache directory (created if missing)
    :param size_limit_bytes: total cache budget
    :param expected_row_size_bytes: used only for the shard-capacity sanity
        check, mirroring reference local_disk_cache.py:47-49
    :param shards: number of shard subdirectories
    :param cleanup: when True, :meth:`cleanup` removes the entire directory
        (reference local_disk_cache.py:69-83)
    """

    def __init__(self, path, size_limit_bytes, expected_row_size_bytes=0,
                 shards=6, cleanup=False):
        self._path = path
        self._shards = shards
        self._shard_limit = size_limit_bytes // shards
        self._cleanup = cleanup
        self._lock = threading.Lock()
        if expected_row_size_bytes and self._shard_limit < 5 * expected_row_size_bytes:
            raise ValueError(
                'Condition (size_limit_bytes/shards) >= 5*expected_row_size_bytes '
                'does not hold: per-shard budget {} is too small for rows of {} '
                'bytes'.format(self._shard_limit, expected_row_size_bytes))
        for i in range(shards):
            os.makedirs(self._shard_dir(i), exist_ok=True)

    def _shard_dir(self, i):
        return os.path.join(self._path, 'shard-{:02d}'.format(i))

    def _entry_path(self, key):
        digest = hashlib.sha1(str(key).encode('utf-8')).hexdigest()
        shard = int(digest[:8], 16) % self._shards
        return self._shard_dir(shard), os.path.join(
            self._shard_dir(shard), digest + '.pkl')

    def get(self, key, fill_cache_func):
        shard_dir, path = self._entry_path(key)
        try:
            with open(path, 'rb') as f:
                return pickle.load(f)
        except (OSError, pickle.UnpicklingError, EOFError, ValueError,
                AttributeError, IndexError, ImportError):
            # a corrupt/truncated entry (partial write, version skew) is a
            # miss: refill and overwrite
            pass
        value = fill_cache_func()
        data = pickle.dumps(value, protocol=pickle.HIGHEST_PROTOCOL)
        with self._lock:
            self._evict_for(shard_dir, len(data))
            tmp = path + '.tmp.{}'.format(os.getpid())
            with open(tmp, 'wb') as f:
                f.write(data)
            os.replace(tmp, path)
        return value

    def _evict_for(self, shard_dir, incoming_bytes):
        """Least-recently-stored eviction within one shard."""
        entries = []
        total = 0
        for name in os.listdir(shard_dir):
            p = os.path.join(shard_dir, name)
            try:
                st = os.stat(p)
            except OSError:
                continue
            entries.append((st.st_mtime, st.st_size, p))
            total += st.st_size
        entries.sort()
        while entries and total + incoming_bytes > self._shard_limit:
            _, size, p = entries.pop(0)
            try:
                os.remove(p)
            except OSError:
                pass
            total -= size

    def size_bytes(self):
        total = 0
        for i in range(self._shards):
            d = self._shard_dir(i)
            for name in os.listdir(d):
                try:
                    total += os.stat(os.path.join(d, name)).st_size
                except OSError:
                    pass
        return total

    def cleanup(self):
        if self._cleanup:
            shutil.rmtree(self._path, ignore_errors=True)
