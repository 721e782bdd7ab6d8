"""FIFO of column batches re-chunked into fixed-size batches.

Parity: /root/reference/petastorm/pyarrow_helpers/batching_table_queue.py
(:20-79 BatchingTableQueue) — the reference version re-chunks Arrow tables;
this framework's unit is the numpy/torch column-dict, so the queue operates
on those (it is the building block BatchedNoopShufflingBuffer uses on the
loader path; provided standalone for parity and reuse).
"""

from collections import deque

import numpy as np


def _concat(parts):
    try:
        import torch
        if parts and isinstance(next(iter(parts[0].values())), torch.Tensor):
            return {k: torch.cat([p[k] for p in parts]) for k in parts[0]}
    except ImportError:  # pragma: no cover
        pass
    return {k: np.concatenate([p[k] for p in parts]) for k in parts[0]}


class BatchingQueue(object):
    """put() column-dicts of any length; get() returns exactly
    ``batch_size`` rows (None until enough rows are buffered)."""

    def __init__(self, batch_size):
        self._batch_size = batch_size
        self._chunks = deque()
        self._size = 0

    def put(self, columns):
        n = len(next(iter(columns.values()))) if columns else 0
        if n:
            self._chunks.append(columns)
            self._size += n

    def empty(self):
        return self._size < self._batch_size

    @property
    def size(self):
        return self._size

    def get(self):
        if self.empty():
            return None
        want = self._batch_size
        parts = []
        got = 0
        while got < want:
            chunk = self._chunks[0]
            n = len(next(iter(chunk.values())))
            take = min(n, want - got)
            if take == n:
                parts.append(self._chunks.popleft())
            else:
                parts.append({k: v[:take] for k, v in chunk.items()})
                self._chunks[0] = {k: v[take:] for k, v in chunk.items()}
            got += take
        self._size -= got
        return parts[0] if len(parts) == 1 else _concat(parts)
