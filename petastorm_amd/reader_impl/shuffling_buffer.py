"""Shuffling buffers: bounded reservoirs that decouple reader order from
delivery order.

Parity: /root/reference/petastorm/reader_impl/shuffling_buffer.py
(NoopShufflingBuffer :75-100, RandomShufflingBuffer :103-180 with the
swap-with-last O(1) retrieve :158-167) and
pytorch_shuffling_buffer.py (batched torch-tensor variants :85-279).

The batched variants store whole column tensors and slice batches out of a
presampled permutation — on the GPU path the tensors live in HBM and the
gather is the ``gpu_shuffle_gather`` HIP kernel (torch index_select on
device), mirroring reference pytorch_shuffling_buffer.py:252-266.
"""

from collections import deque

import numpy as np


class ShufflingBufferBase(object):
    def add_many(self, items):
        raise NotImplementedError()

    def retrieve(self):
        raise NotImplementedError()

    def can_add(self):
        raise NotImplementedError()

    def can_retrieve(self):
        raise NotImplementedError()

    @property
    def size(self):
        raise NotImplementedError()

    def finish(self):
        """No more items will be added; drain whatever remains."""
        raise NotImplementedError()


class NoopShufflingBuffer(ShufflingBufferBase):
    """FIFO pass-through (reference shuffling_buffer.py:75-100)."""

    def __init__(self):
        self._q = deque()
        self._done = False

    def add_many(self, items):
        self._q.extend(items)

    def retrieve(self):
        return self._q.popleft()

    def can_add(self):
        return not self._done

    def can_retrieve(self):
        return len(self._q) > 0

    @property
    def size(self):
        return len(self._q)

    def finish(self):
        self._done = True


class RandomShufflingBuffer(ShufflingBufferBase):
    """Uniform random retrieve with O(1) swap-with-last removal
    (reference shuffling_buffer.py:103-180).

    :param shuffling_buffer_capacity: soft capacity; ``can_add`` is False
        once size reaches it
    :param min_after_retrieve: retrieval allowed only while
        ``size >= min_after_retrieve`` (until :meth:`finish`), which
        guarantees a minimum mixing pool
    :param extra_capacity: headroom for multi-item adds beyond capacity
    """

    def __init__(self, shuffling_buffer_capacity, min_after_retrieve,
                 extra_capacity=1000, seed=None):
        if min_after_retrieve >= shuffling_buffer_capacity:
            raise ValueError('min_after_retrieve must be smaller than '
                             'shuffling_buffer_capacity')
        self._capacity = shuffling_buffer_capacity
        self._min_after_retrieve = min_after_retrieve
        self._extra_capacity = extra_capacity
        self._items = []
        self._done = False
        self._rng = np.random.RandomState(seed)

    def add_many(self, items):
        if self._done:
            raise RuntimeError('Can not add to a finished buffer')
        if not self.can_add():
            raise RuntimeError('Buffer is over capacity; check can_add() first')
        self._items.extend(items)
        if len(self._items) > self._capacity + self._extra_capacity:
            raise RuntimeError('Buffer exceeded capacity+extra_capacity: '
                               'add_many batch too large')

    def retrieve(self):
        if not self.can_retrieve():
            raise RuntimeError('Can not retrieve; check can_retrieve() first')
        idx = int(self._rng.randint(0, len(self._items)))
        # O(1) removal: swap with last then pop (reference :158-167)
        self._items[idx], self._items[-1] = self._items[-1], self._items[idx]
        return self._items.pop()

    def can_add(self):
        return len(self._items) < self._capacity and not self._done

    def can_retrieve(self):
        if self._done:
            return len(self._items) > 0
        return len(self._items) >= self._min_after_retrieve

    @property
    def size(self):
        return len(self._items)

    def finish(self):
        self._done = True


# ---------------------------------------------------------------------------
# batched (torch) variants
# ---------------------------------------------------------------------------

class BatchedNoopShufflingBuffer(ShufflingBufferBase):
    """FIFO over column-tensor chunks, slicing fixed-size batches
    (reference pytorch_shuffling_buffer.py:85-134)."""

    def __init__(self, batch_size):
        self._batch_size = batch_size
        self._chunks = deque()   # each: dict name -> tensor
        self._size = 0
        self._done = False

    def add_many(self, columns):
        n = _num_rows(columns)
        if n:
            self._chunks.append(columns)
            self._size += n

    def retrieve(self):
        import torch
        want = self._batch_size if not self._done \
            else min(self._batch_size, self._size)
        got = 0
        parts = []
        while got < want and self._chunks:
            chunk = self._chunks[0]
            n = _num_rows(chunk)
            take = min(n, want - got)
            if take == n:
                parts.append(self._chunks.popleft())
            else:
                parts.append({k: v[:take] for k, v in chunk.items()})
                self._chunks[0] = {k: v[take:] for k, v in chunk.items()}
            got += take
        self._size -= got
        if len(parts) == 1:
            return parts[0]
        return {k: torch.cat([p[k] for p in parts]) for k in parts[0]}

    def can_add(self):
        return not self._done

    def can_retrieve(self):
        return self._size >= self._batch_size or (self._done and self._size > 0)

    @property
    def size(self):
        return self._size

    def finish(self):
        self._done = True


class BatchedRandomShufflingBuffer(ShufflingBufferBase):
    """Column-tensor shuffling pool over a PREALLOCATED buffer.

    Semantics follow reference pytorch_shuffling_buffer.py:137-279 (random
    batches out of a bounded mixing pool), but the implementation is the
    vectorized analog of the reference's O(1) swap-with-last single-row
    buffer (reference shuffling_buffer.py:158-167): a fixed
    ``[capacity + extra, ...]`` tensor per column, appends copy into free
    slots, and a retrieve gathers ``batch_size`` random rows then back-fills
    their slots from the tail.  Per-operation cost is O(rows moved) — the
    previous implementation re-``torch.cat``-ed the whole pool on every add.

    Works for CPU and CUDA tensors alike — on CUDA the gathers/scatters stay
    HBM-resident.
    """

    def __init__(self, shuffling_buffer_capacity, min_after_retrieve,
                 batch_size, extra_capacity=100000, seed=None):
        if min_after_retrieve >= shuffling_buffer_capacity:
            raise ValueError('min_after_retrieve must be smaller than '
                             'shuffling_buffer_capacity')
        self._capacity = shuffling_buffer_capacity
        self._min_after_retrieve = min_after_retrieve
        self._batch_size = batch_size
        self._extra_capacity = extra_capacity
        self._pool = None           # name -> [cap_total, ...] tensor
        self._cap_total = None
        self._size = 0
        self._done = False
        self._generator = None
        self._seed = seed

    def _torch_gen(self, device):
        import torch
        if self._generator is None and self._seed is not None:
            self._generator = torch.Generator(device=device)
            self._generator.manual_seed(self._seed)
        return self._generator

    def add_many(self, columns):
        import torch
        n = _num_rows(columns)
        if n == 0:
            return
        if self._size + n > self._capacity + self._extra_capacity:
            raise RuntimeError('Buffer exceeded capacity+extra_capacity')
        if self._pool is None:
            # size for capacity + one add-chunk of headroom; grows on demand
            # (extra_capacity is an upper BOUND, not a preallocation)
            self._cap_total = self._capacity + n
            self._pool = {
                k: torch.empty((self._cap_total,) + tuple(v.shape[1:]),
                               dtype=v.dtype, device=v.device)
                for k, v in columns.items()}
        elif self._size + n > self._cap_total:
            new_cap = min(self._capacity + self._extra_capacity,
                          max(self._cap_total * 2, self._size + n))
            for k, v in self._pool.items():
                grown = torch.empty((new_cap,) + tuple(v.shape[1:]),
                                    dtype=v.dtype, device=v.device)
                grown[:self._size] = v[:self._size]
                self._pool[k] = grown
            self._cap_total = new_cap
        for k, v in columns.items():
            self._pool[k][self._size:self._size + n] = v
        self._size += n

    def retrieve(self):
        import torch
        want = min(self._batch_size, self._size)
        device = next(iter(self._pool.values())).device
        perm = torch.randperm(self._size, device=device,
                              generator=self._torch_gen(device))
        batch_idx = perm[:want]
        batch = {k: v.index_select(0, batch_idx)
                 for k, v in self._pool.items()}
        # back-fill the removed slots from the tail (vectorized
        # swap-with-last).  A tail row chosen for the batch would be
        # clobbered by the scatter, so only relocate tail rows that SURVIVE,
        # into surviving holes below the new size.
        new_size = self._size - want
        in_tail = batch_idx >= new_size
        holes = batch_idx[~in_tail]                  # empty slots below cut
        if holes.numel():
            # surviving tail rows = tail slots NOT picked into the batch;
            # their count equals the number of holes by construction
            tail = torch.arange(new_size, self._size, device=device)
            picked_tail = batch_idx[in_tail] - new_size
            tail_mask = torch.ones(want, dtype=torch.bool, device=device)
            tail_mask[picked_tail] = False
            movers = tail[tail_mask]
            for k, v in self._pool.items():
                v[holes] = v.index_select(0, movers)
        self._size = new_size
        return batch

    def can_add(self):
        return self._size < self._capacity and not self._done

    def can_retrieve(self):
        if self._done:
            return self._size > 0
        return self._size >= max(self._min_after_retrieve, self._batch_size)

    @property
    def size(self):
        return self._size

    def finish(self):
        self._done = True


def _num_rows(columns):
    if not columns:
        return 0
    return len(next(iter(columns.values())))
