"""Property-based round-trip tests (hypothesis): codecs and schema JSON.

The reference pins codec behavior with example-based tests
(tests/test_codec_*); property testing sweeps the input space (shapes,
dtypes, values incl. NaN/inf/extremes) for the encode->decode contract.
"""
from decimal import Decimal

import numpy as np
from hypothesis import given, settings
from hypothesis import strategies as st
from hypothesis.extra import numpy as hnp

from petastorm_amd.codecs import (CompressedNdarrayCodec, NdarrayCodec,
                                  ScalarCodec)
from petastorm_amd.unischema import Unischema, UnischemaField

_DTYPES = st.sampled_from([np.uint8, np.int16, np.int32, np.int64,
                           np.float32, np.float64])


@st.composite
def arrays(draw):
    dtype = draw(_DTYPES)
    shape = draw(st.lists(st.integers(1, 6), min_size=1, max_size=3))
    return draw(hnp.arrays(dtype, tuple(shape)))


@settings(max_examples=60, deadline=None)
@given(arr=arrays())
def test_ndarray_codec_roundtrip(arr):
    field = UnischemaField('x', arr.dtype.type, tuple(arr.shape),
                           NdarrayCodec(), False)
    out = field.codec.decode(field, field.codec.encode(field, arr))
    assert out.dtype == arr.dtype and out.shape == arr.shape
    np.testing.assert_array_equal(out, arr)


@settings(max_examples=40, deadline=None)
@given(arr=arrays(), level=st.integers(1, 9))
def test_compressed_ndarray_codec_roundtrip(arr, level):
    field = UnischemaField('x', arr.dtype.type, tuple(arr.shape),
                           CompressedNdarrayCodec(level), False)
    out = field.codec.decode(field, field.codec.encode(field, arr))
    np.testing.assert_array_equal(out, arr)


@settings(max_examples=60, deadline=None)
@given(v=st.one_of(
    st.integers(-2**62, 2**62).map(np.int64),
    st.floats(allow_nan=False, width=64).map(np.float64),
    st.booleans().map(np.bool_),
    st.text(max_size=40),
))
def test_scalar_codec_roundtrip(v):
    np_dtype = type(v) if isinstance(v, np.generic) else np.str_
    field = UnischemaField('s', np_dtype, (), ScalarCodec(), False)
    out = field.codec.decode(field, field.codec.encode(field, v))
    if isinstance(v, np.floating):
        assert out == v or (np.isnan(out) and np.isnan(v))
    else:
        assert out == v


@settings(max_examples=40, deadline=None)
@given(digits=st.integers(0, 10**24), scale=st.integers(0, 12),
       neg=st.booleans())
def test_scalar_codec_decimal_exact(digits, scale, neg):
    v = Decimal(digits) / (Decimal(10) ** scale)
    if neg:
        v = -v
    field = UnischemaField('d', Decimal, (), ScalarCodec(), False)
    out = field.codec.decode(field, field.codec.encode(field, v))
    assert isinstance(out, Decimal) and out == v


@settings(max_examples=30, deadline=None)
@given(names=st.lists(
    st.from_regex(r'[a-z][a-z0-9_]{0,12}', fullmatch=True),
    min_size=1, max_size=8, unique=True))
def test_unischema_json_roundtrip(names):
    fields = [UnischemaField(n, np.int64, (), ScalarCodec(), False)
              for n in names]
    schema = Unischema('P', fields)
    back = Unischema.from_json(schema.to_json())
    assert list(back.fields) == list(schema.fields)
    for n in names:
        f = back.fields[n]
        assert f.numpy_dtype is np.int64 and f.shape == ()
