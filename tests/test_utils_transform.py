"""Direct unit coverage for utils.decode_row / run_in_subprocess and
transform.transform_schema (indirectly exercised everywhere; pinned here
so semantics changes fail loudly).

Parity: reference petastorm/utils.py:52-85 (decode_row + DecodeFieldError),
petastorm/utils.py:28-45 (run_in_subprocess),
petastorm/transform.py:60-89 (transform_schema edit/remove semantics).
"""
import numpy as np
import pytest

from petastorm_amd.codecs import NdarrayCodec, ScalarCodec
from petastorm_amd.transform import TransformSpec, transform_schema
from petastorm_amd.unischema import Unischema, UnischemaField
from petastorm_amd.errors import DecodeFieldError
from petastorm_amd.utils import decode_row, run_in_subprocess


SCHEMA = Unischema('S', [
    UnischemaField('id', np.int32, (), ScalarCodec(), False),
    UnischemaField('vec', np.float32, (4,), NdarrayCodec(), True),
])


def test_decode_row_codec_and_null():
    enc_vec = NdarrayCodec().encode(SCHEMA.fields['vec'],
                                    np.arange(4, dtype=np.float32))
    row = decode_row({'id': 7, 'vec': enc_vec}, SCHEMA)
    assert row['id'] == 7
    np.testing.assert_array_equal(row['vec'],
                                  np.arange(4, dtype=np.float32))
    # nullable field: None passes through undecoded
    row = decode_row({'id': 1, 'vec': None}, SCHEMA)
    assert row['vec'] is None


def test_decode_row_wraps_field_errors():
    with pytest.raises(DecodeFieldError):
        decode_row({'id': 1, 'vec': b'not-an-npy-payload'}, SCHEMA)


def _sub(a, b):
    return a * b


def test_run_in_subprocess():
    assert run_in_subprocess(_sub, 6, 7) == 42


def test_transform_schema_edit_remove():
    ts = TransformSpec(
        func=None,
        edit_fields=[UnischemaField('vec', np.float64, (2, 2), None, False)],
        removed_fields=['id'])
    out = transform_schema(SCHEMA, ts)
    assert set(out.fields) == {'vec'}
    assert out.fields['vec'].numpy_dtype == np.float64
    assert out.fields['vec'].shape == (2, 2)


def test_transform_schema_selected_fields():
    ts = TransformSpec(func=None, selected_fields=['id'])
    out = transform_schema(SCHEMA, ts)
    assert set(out.fields) == {'id'}
