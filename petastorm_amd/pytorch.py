"""PyTorch loader adapters over a Reader.

Parity: /root/reference/petastorm/pytorch.py.

* ``_sanitize_pytorch_types`` dtype promotion (reference :40-70):
  uint16 -> int32, uint32 -> int64, bool -> uint8
* ``decimal_friendly_collate`` (reference :73-95)
* ``DataLoader``: per-row accumulation + collate, optional shuffling queue
  (reference :175-248); batched readers are transposed column->rows
  (reference :207-216)
* ``BatchedDataLoader``: column-tensor shuffling buffers, batched slicing
  (reference :259-370)
* ``InMemBatchedDataLoader``: loads <= capacity rows once, epoch shuffles via
  ``torch.randperm(seed+epoch)`` (reference :437-501)

MI355X note: when the underlying reader is the GPU pipeline the columns are
already CUDA tensors; ``BatchedDataLoader`` then shuffles/slices entirely in
HBM (zero host round-trips).
"""

from decimal import Decimal

import numpy as np

from petastorm_amd.reader_impl.shuffling_buffer import (
    BatchedNoopShufflingBuffer, BatchedRandomShufflingBuffer,
    NoopShufflingBuffer, RandomShufflingBuffer)

_PROMOTIONS = {
    'uint16': np.int32,
    'uint32': np.int64,
    'bool': np.uint8,
}


def _sanitize_pytorch_types(row_as_dict):
    """In-place dtype promotion for types torch can't represent
    (reference pytorch.py:40-70)."""
    for name, value in row_as_dict.items():
        if isinstance(value, np.ndarray):
            kind = value.dtype.name
            if kind in _PROMOTIONS:
                row_as_dict[name] = value.astype(_PROMOTIONS[kind])
            elif value.dtype.kind == 'U':
                pass  # strings collate as python lists
        elif isinstance(value, np.number):
            kind = value.dtype.name
            if kind in _PROMOTIONS:
                row_as_dict[name] = _PROMOTIONS[kind](value)


def decimal_friendly_collate(batch):
    """Like torch's default_collate but Decimals collate to str lists
    (reference pytorch.py:73-95)."""
    import torch
    from torch.utils.data._utils.collate import default_collate
    if isinstance(batch[0], Decimal):
        return [str(v) for v in batch]
    if isinstance(batch[0], dict):
        return {k: decimal_friendly_collate([r[k] for r in batch])
                for k in batch[0]}
    if isinstance(batch[0], str):
        return list(batch)
    if isinstance(batch[0], (list, tuple)):
        transposed = list(zip(*batch))
        out = [decimal_friendly_collate(list(s)) for s in transposed]
        return type(batch[0])(out) if not isinstance(batch[0], tuple) \
            else tuple(out)
    return default_collate(batch)


class LoaderBase(object):
    """Iteration guard + auto reader.reset (reference pytorch.py:103-128)."""

    def __init__(self, reader):
        self.reader = reader
        self._in_iter = None

    def __iter__(self):
        if self._in_iter is not None and self._in_iter:
            raise RuntimeError('Only one iteration over the loader may be '
                               'active at a time')
        if self._in_iter is not None:
            # second epoch over the same loader: reset the reader
            self.reader.reset()
        self._in_iter = True
        try:
            for batch in self._iter_impl():
                yield batch
        finally:
            self._in_iter = False

    def _iter_impl(self):
        raise NotImplementedError()

    def stop(self):
        self.reader.stop()

    def join(self):
        self.reader.join()

    def __enter__(self):
        return self

    def __exit__(self, exc_type, exc_val, exc_tb):
        self.stop()
        self.join()


class DataLoader(LoaderBase):
    """Row-collating loader (reference pytorch.py:131-248)."""

    def __init__(self, reader, batch_size=1,
                 collate_fn=decimal_friendly_collate,
                 shuffling_queue_capacity=0, seed=None):
        super(DataLoader, self).__init__(reader)
        self.batch_size = batch_size
        self.collate_fn = collate_fn
        self.shuffling_queue_capacity = shuffling_queue_capacity
        self._seed = seed

    def _iter_impl(self):
        if self.shuffling_queue_capacity > 0:
            min_after = max(1, self.shuffling_queue_capacity // 2)
            buffer = RandomShufflingBuffer(
                self.shuffling_queue_capacity, min_after,
                extra_capacity=100000, seed=self._seed)
        else:
            buffer = NoopShufflingBuffer()
        batch_acc = []
        for row in self.reader:
            if self.reader.ngram is not None:
                rows = [{ts: nt._asdict() for ts, nt in row.items()}]
                # sanitize each timestep dict
                for r in rows:
                    for ts in r:
                        _sanitize_pytorch_types(r[ts])
            elif self.reader.batched_output:
                # transpose columns into row dicts (reference :207-216)
                cols = row._asdict()
                names = list(cols.keys())
                n = len(cols[names[0]])
                rows = []
                for i in range(n):
                    d = {k: cols[k][i] for k in names}
                    _sanitize_pytorch_types(d)
                    rows.append(d)
            else:
                d = row._asdict()
                _sanitize_pytorch_types(d)
                rows = [d]
            while not buffer.can_add() and buffer.can_retrieve():
                batch_acc.append(buffer.retrieve())
                if len(batch_acc) == self.batch_size:
                    yield self.collate_fn(batch_acc)
                    batch_acc = []
            buffer.add_many(rows)
            while buffer.can_retrieve():
                batch_acc.append(buffer.retrieve())
                if len(batch_acc) == self.batch_size:
                    yield self.collate_fn(batch_acc)
                    batch_acc = []
        buffer.finish()
        while buffer.can_retrieve():
            batch_acc.append(buffer.retrieve())
            if len(batch_acc) == self.batch_size:
                yield self.collate_fn(batch_acc)
                batch_acc = []
        if batch_acc:
            # partial final batch (reference :231-233)
            yield self.collate_fn(batch_acc)


def _columns_to_tensors(columns, transform_fn):
    import torch
    out = {}
    for k, v in columns.items():
        if isinstance(v, torch.Tensor):
            out[k] = v
            continue
        if isinstance(v, np.ndarray):
            if v.dtype.kind in 'OUS':
                raise TypeError(
                    'Field {!r} has non-numeric dtype {}; BatchedDataLoader '
                    'requires numeric columns (reference pytorch.py:294). '
                    'Remove it with schema_fields/TransformSpec or use '
                    'DataLoader.'.format(k, v.dtype))
            d = dict([(k, v)])
            _sanitize_pytorch_types(d)
            v = d[k]
        out[k] = transform_fn(v)
    return out


class BatchedDataLoader(LoaderBase):
    """Column-tensor loader: much faster at large batch sizes
    (reference pytorch.py:259-370)."""

    def __init__(self, reader, batch_size=1, transform_fn=None,
                 shuffling_queue_capacity=0, seed=None):
        super(BatchedDataLoader, self).__init__(reader)
        if reader.ngram is not None:
            raise NotImplementedError('BatchedDataLoader does not support '
                                      'NGram readers')
        self.batch_size = batch_size
        import torch
        self.transform_fn = transform_fn or torch.as_tensor
        self.shuffling_queue_capacity = shuffling_queue_capacity
        self._seed = seed

    def _iter_impl(self):
        if self.shuffling_queue_capacity > 0:
            min_after = max(1, self.shuffling_queue_capacity // 2)
            buffer = BatchedRandomShufflingBuffer(
                self.shuffling_queue_capacity, min_after, self.batch_size,
                seed=self._seed)
        else:
            buffer = BatchedNoopShufflingBuffer(self.batch_size)
        keys = None
        for item in self.reader:
            if self.reader.batched_output:
                columns = item._asdict()
            else:
                d = item._asdict()
                columns = {k: np.asarray([v]) for k, v in d.items()}
            columns = _columns_to_tensors(columns, self.transform_fn)
            if keys is None:
                keys = list(columns.keys())
            while not buffer.can_add() and buffer.can_retrieve():
                yield self._emit(buffer.retrieve(), keys)
            buffer.add_many(columns)
            while buffer.can_retrieve():
                yield self._emit(buffer.retrieve(), keys)
        buffer.finish()
        while buffer.can_retrieve():
            yield self._emit(buffer.retrieve(), keys)

    @staticmethod
    def _emit(batch, keys):
        return {k: batch[k] for k in keys}


class InMemBatchedDataLoader(LoaderBase):
    """Loads up to ``rows_capacity`` rows once; serves ``num_epochs`` epochs
    of shuffled batches from memory (reference pytorch.py:437-501)."""

    def __init__(self, reader, batch_size=1, transform_fn=None,
                 num_epochs=1, rows_capacity=1024, shuffle=True, seed=0):
        super(InMemBatchedDataLoader, self).__init__(reader)
        import torch
        self.batch_size = batch_size
        self.transform_fn = transform_fn or torch.as_tensor
        self._num_epochs = num_epochs
        self._capacity = rows_capacity
        self._shuffle = shuffle
        self._seed = seed
        self._epoch = 0
        self._columns = None

    def _load_once(self):
        import torch
        if self._columns is not None:
            return
        chunks = []
        loaded = 0
        for item in self.reader:
            if self.reader.batched_output:
                columns = item._asdict()
            else:
                columns = {k: np.asarray([v]) for k, v in item._asdict().items()}
            columns = _columns_to_tensors(columns, self.transform_fn)
            n = len(next(iter(columns.values())))
            take = min(n, self._capacity - loaded)
            if take < n:
                columns = {k: v[:take] for k, v in columns.items()}
            chunks.append(columns)
            loaded += take
            if loaded >= self._capacity:
                break
        if not chunks:
            raise RuntimeError('Reader produced no rows')
        self._columns = {k: torch.cat([c[k] for c in chunks])
                         for k in chunks[0]}

    def __iter__(self):
        # overrides LoaderBase: the reader is consumed exactly once
        if self._epoch >= self._num_epochs:
            raise RuntimeError('InMemBatchedDataLoader: all {} epochs '
                               'consumed'.format(self._num_epochs))
        self._load_once()
        return self._epoch_iter()

    def _epoch_iter(self):
        import torch
        n = len(next(iter(self._columns.values())))
        if self._shuffle:
            g = torch.Generator()
            g.manual_seed(self._seed + self._epoch)  # reference :479-485
            order = torch.randperm(n, generator=g)
        else:
            order = torch.arange(n)
        self._epoch += 1
        for lo in range(0, n, self.batch_size):
            idx = order[lo:lo + self.batch_size]
            yield {k: v[idx] for k, v in self._columns.items()}
