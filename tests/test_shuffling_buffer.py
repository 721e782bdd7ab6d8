"""Shuffling buffers (parity: reference tests/test_shuffling_buffer.py)."""
import numpy as np
import pytest
import torch

from petastorm_amd.reader_impl.shuffling_buffer import (
    BatchedNoopShufflingBuffer, BatchedRandomShufflingBuffer,
    NoopShufflingBuffer, RandomShufflingBuffer)


def test_noop_fifo():
    b = NoopShufflingBuffer()
    b.add_many([1, 2, 3])
    assert b.size == 3
    assert [b.retrieve() for _ in range(3)] == [1, 2, 3]


def test_random_min_after_retrieve():
    b = RandomShufflingBuffer(10, min_after_retrieve=3, seed=0)
    b.add_many([1, 2])
    assert not b.can_retrieve()
    b.add_many([3, 4])
    assert b.can_retrieve()
    b.retrieve()
    b.retrieve()  # size 2 < 3
    assert not b.can_retrieve()
    b.finish()
    assert b.can_retrieve()
    got = {b.retrieve(), b.retrieve()}
    assert got <= {1, 2, 3, 4}
    assert not b.can_retrieve()


def test_random_capacity_gate():
    b = RandomShufflingBuffer(3, min_after_retrieve=1, seed=0)
    b.add_many([1, 2, 3])
    assert not b.can_add()
    b.retrieve()
    assert b.can_add()


def test_random_yields_all_items():
    b = RandomShufflingBuffer(100, min_after_retrieve=10, seed=42)
    items = list(range(50))
    b.add_many(items)
    b.finish()
    out = []
    while b.can_retrieve():
        out.append(b.retrieve())
    assert sorted(out) == items
    assert out != items  # shuffled with high probability


def test_random_invalid_params():
    with pytest.raises(ValueError):
        RandomShufflingBuffer(5, min_after_retrieve=5)


def _cols(lo, hi):
    return {'x': torch.arange(lo, hi), 'y': torch.arange(lo, hi) * 10}


def test_batched_noop_slices_batches():
    b = BatchedNoopShufflingBuffer(batch_size=4)
    b.add_many(_cols(0, 6))
    b.add_many(_cols(6, 10))
    assert b.can_retrieve()
    batch = b.retrieve()
    assert torch.equal(batch['x'], torch.arange(0, 4))
    batch2 = b.retrieve()
    assert torch.equal(batch2['x'], torch.arange(4, 8))
    assert not b.can_retrieve()  # only 2 rows left < batch
    b.finish()
    batch3 = b.retrieve()
    assert torch.equal(batch3['x'], torch.arange(8, 10))


def test_batched_random_covers_all_rows():
    b = BatchedRandomShufflingBuffer(100, min_after_retrieve=10,
                                     batch_size=8, seed=1)
    b.add_many(_cols(0, 64))
    b.finish()
    xs = []
    while b.can_retrieve():
        batch = b.retrieve()
        assert torch.equal(batch['y'], batch['x'] * 10)  # row integrity
        xs.append(batch['x'])
    allx = torch.cat(xs)
    assert sorted(allx.tolist()) == list(range(64))
    assert allx.tolist() != list(range(64))


def test_batched_random_min_after():
    b = BatchedRandomShufflingBuffer(100, min_after_retrieve=32,
                                     batch_size=8, seed=1)
    b.add_many(_cols(0, 16))
    assert not b.can_retrieve()
    b.add_many(_cols(16, 48))
    assert b.can_retrieve()


# ---------------------------------------------------------------------------
# BatchingQueue (reference pyarrow_helpers/batching_table_queue.py:20-79)
# ---------------------------------------------------------------------------

def test_batching_queue_rechunks_exact_batches():
    import numpy as np
    from petastorm_amd.reader_impl.batching_queue import BatchingQueue
    q = BatchingQueue(batch_size=10)
    total = 0
    for n in (3, 9, 25, 2, 14):  # uneven puts
        q.put({'a': np.arange(total, total + n),
               'b': np.arange(total, total + n) * 2.0})
        total += n
    got = []
    while not q.empty():
        b = q.get()
        assert len(b['a']) == 10
        np.testing.assert_array_equal(b['b'], b['a'] * 2.0)
        got.extend(b['a'].tolist())
    assert got == list(range((total // 10) * 10))
    assert q.size == total % 10
    assert q.get() is None  # not enough rows buffered


def test_batching_queue_torch_tensors():
    import numpy as np
    import torch
    from petastorm_amd.reader_impl.batching_queue import BatchingQueue
    q = BatchingQueue(batch_size=4)
    q.put({'x': torch.arange(3)})
    q.put({'x': torch.arange(3, 9)})
    b = q.get()
    assert isinstance(b['x'], torch.Tensor)
    np.testing.assert_array_equal(b['x'].numpy(), [0, 1, 2, 3])
