"""HBM-resident cache of decoded row-groups.

The MI355X analogue of the reference's LocalDiskCache
(petastorm/local_disk_cache.py): with 288 GB of HBM3E per GPU, decoded
row-groups are cached *on device* so later epochs skip disk IO and decode
entirely.  Eviction is LRU by insertion/access order within a byte budget.
"""

from collections import OrderedDict

from petastorm_amd.cache import CacheBase


def _nbytes(value):
    import torch
    total = 0
    if isinstance(value, dict):
        items = value.values()
    else:
        items = [value]
    for v in items:
        if isinstance(v, torch.Tensor):
            total += v.numel() * v.element_size()
        elif hasattr(v, 'nbytes'):
            total += int(v.nbytes)
    return total


class HbmCache(CacheBase):
    def __init__(self, size_limit_bytes):
        self._limit = size_limit_bytes
        self._store = OrderedDict()
        self._bytes = 0
        self.hits = 0
        self.misses = 0

    def get(self, key, fill_cache_func):
        if key in self._store:
            self.hits += 1
            self._store.move_to_end(key)
            value = self._store[key]
            # multi-stream readers gather from cached tensors on their own
            # stream: mark that use so a later LRU eviction cannot hand the
            # blocks back to the allocator while the gather is in flight
            import torch
            if torch.cuda.is_available():
                cur = torch.cuda.current_stream()
                for v in (value.values() if isinstance(value, dict)
                          else [value]):
                    if isinstance(v, torch.Tensor) and v.is_cuda:
                        v.record_stream(cur)
            return value
        self.misses += 1
        value = fill_cache_func()
        nb = _nbytes(value)
        if nb <= self._limit:
            while self._bytes + nb > self._limit and self._store:
                _, old = self._store.popitem(last=False)
                self._bytes -= _nbytes(old)
            self._store[key] = value
            self._bytes += nb
        return value

    @property
    def size_bytes(self):
        return self._bytes

    def cleanup(self):
        self._store.clear()
        self._bytes = 0
