"""Batched NGram windowing: torch implementation vs the CPU source of truth.

These run on CPU here (torch ops are device-agnostic); the GPU suite
re-exercises them on-device via test_gpu_decode.py's reader tests.
"""
import numpy as np
import torch

from petastorm_amd.codecs import ScalarCodec
from petastorm_amd.gpu.ngram import form_ngram_batched, window_starts
from petastorm_amd.ngram import NGram
from petastorm_amd.unischema import Unischema, UnischemaField

Schema = Unischema('S', [
    UnischemaField('ts', np.int64, (), ScalarCodec(), False),
    UnischemaField('a', np.int32, (), ScalarCodec(), False),
])


def _ngram(length=3, delta=2, overlap=True):
    fields = {i: [Schema.ts, Schema.a] for i in range(length)}
    return NGram(fields, delta_threshold=delta, timestamp_field=Schema.ts,
                 timestamp_overlap=overlap)


def test_window_starts_matches_cpu_reference():
    rng = np.random.RandomState(0)
    for trial in range(20):
        n = rng.randint(2, 200)
        ts = np.cumsum(rng.randint(1, 5, n)).astype(np.int64)
        for overlap in (True, False):
            ng = _ngram(length=rng.randint(2, 5), delta=rng.randint(1, 4),
                        overlap=overlap)
            order, exp = ng.form_ngram_indices(ts)
            got = window_starts(torch.from_numpy(ts[order]), ng.length,
                                ng.delta_threshold, overlap)
            np.testing.assert_array_equal(got.numpy(), exp,
                                          err_msg='trial {}'.format(trial))


def test_form_ngram_batched_values():
    ts = np.array([0, 1, 2, 6, 7, 8, 20], dtype=np.int64)
    a = (ts * 10).astype(np.int32)
    ng = _ngram(length=2, delta=1)
    out = form_ngram_batched({'ts': torch.from_numpy(ts),
                              'a': torch.from_numpy(a)}, ng)
    # windows: (0,1),(1,2),(6,7),(7,8)
    np.testing.assert_array_equal(out[0]['ts'].numpy(), [0, 1, 6, 7])
    np.testing.assert_array_equal(out[1]['ts'].numpy(), [1, 2, 7, 8])
    np.testing.assert_array_equal(out[0]['a'].numpy(), [0, 10, 60, 70])


def test_form_ngram_batched_matches_row_path():
    rng = np.random.RandomState(1)
    ts = np.cumsum(rng.randint(1, 3, 100)).astype(np.int64)
    a = rng.randint(0, 1000, 100).astype(np.int32)
    rows = [{'ts': int(t), 'a': int(v)} for t, v in zip(ts, a)]
    ng = _ngram(length=3, delta=2)
    row_windows = ng.form_ngram(rows, Schema)
    batched = form_ngram_batched({'ts': torch.from_numpy(ts),
                                  'a': torch.from_numpy(a)}, ng)
    assert len(row_windows) == batched[0]['ts'].numel()
    for i, w in enumerate(row_windows):
        for k in range(3):
            assert w[k].ts == int(batched[k]['ts'][i])
            assert w[k].a == int(batched[k]['a'][i])


def test_unsorted_input_sorted_internally():
    ts = np.array([5, 1, 3, 2, 4], dtype=np.int64)
    a = (ts * 2).astype(np.int32)
    ng = _ngram(length=2, delta=1)
    out = form_ngram_batched({'ts': torch.from_numpy(ts),
                              'a': torch.from_numpy(a)}, ng)
    np.testing.assert_array_equal(out[0]['ts'].numpy(), [1, 2, 3, 4])
