"""NGram assembly (parity: reference tests/test_ngram_end_to_end.py core
semantics, exercised directly against form_ngram)."""
import numpy as np
import pytest

from petastorm_amd.codecs import ScalarCodec
from petastorm_amd.ngram import NGram
from petastorm_amd.unischema import Unischema, UnischemaField

TsSchema = Unischema('TsSchema', [
    UnischemaField('ts', np.int64, (), ScalarCodec(), False),
    UnischemaField('a', np.int32, (), ScalarCodec(), False),
    UnischemaField('b', np.int32, (), ScalarCodec(), False),
])


def _rows(timestamps):
    return [{'ts': t, 'a': 10 * t, 'b': 100 * t} for t in timestamps]


def _make(fields=None, delta=1, overlap=True):
    fields = fields or {0: [TsSchema.ts, TsSchema.a],
                        1: [TsSchema.ts, TsSchema.a]}
    return NGram(fields, delta_threshold=delta, timestamp_field=TsSchema.ts,
                 timestamp_overlap=overlap)


def test_length():
    ng = _make({-1: [TsSchema.a], 0: [TsSchema.a], 1: [TsSchema.a]})
    assert ng.length == 3


def test_non_consecutive_keys_raise():
    with pytest.raises(ValueError):
        NGram({0: [TsSchema.a], 2: [TsSchema.a]}, 1, TsSchema.ts)


def test_basic_windows():
    ng = _make()
    out = ng.form_ngram(_rows([0, 1, 2, 3]), TsSchema)
    assert len(out) == 3
    assert out[0][0].ts == 0 and out[0][1].ts == 1
    assert out[2][0].a == 20


def test_delta_threshold_breaks_window():
    ng = _make(delta=1)
    out = ng.form_ngram(_rows([0, 1, 5, 6]), TsSchema)
    # windows (0,1) and (5,6); (1,5) gap of 4 > 1 excluded
    assert [(w[0].ts, w[1].ts) for w in out] == [(0, 1), (5, 6)]


def test_timestamp_overlap_false():
    ng = _make(overlap=False)
    out = ng.form_ngram(_rows([0, 1, 2, 3]), TsSchema)
    assert [(w[0].ts, w[1].ts) for w in out] == [(0, 1), (2, 3)]


def test_per_timestep_field_selection():
    ng = _make({0: [TsSchema.ts, TsSchema.a], 1: [TsSchema.ts, TsSchema.b]})
    out = ng.form_ngram(_rows([0, 1]), TsSchema)
    assert hasattr(out[0][0], 'a') and not hasattr(out[0][0], 'b')
    assert hasattr(out[0][1], 'b') and not hasattr(out[0][1], 'a')


def test_unsorted_input_is_sorted():
    ng = _make()
    out = ng.form_ngram(_rows([3, 1, 0, 2]), TsSchema)
    assert len(out) == 3 and out[0][0].ts == 0


def test_regex_field_resolution():
    ng = NGram({0: ['t.*', 'a'], 1: ['a']}, 1, TsSchema.ts)
    ng.resolve_regex_field_names(TsSchema)
    assert {f.name for f in ng.fields[0]} == {'ts', 'a'}


def test_form_ngram_indices_matches_form_ngram():
    ng = _make(delta=2)
    ts = [0, 1, 5, 6, 7, 20]
    order, starts = ng.form_ngram_indices(np.array(ts))
    windows = ng.form_ngram(_rows(ts), TsSchema)
    assert len(starts) == len(windows)
    ts_sorted = np.array(sorted(ts))
    for s, w in zip(starts, windows):
        assert ts_sorted[s] == w[0].ts
