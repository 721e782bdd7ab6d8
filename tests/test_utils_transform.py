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


def test_fused_image_normalize_numpy_paths():
    """FusedImageNormalize callable: batch + single-row numpy math matches
    the reference formula, and already-fused input passes through."""
    import numpy as np
    from petastorm_amd.transform import fused_image_normalize
    ts = fused_image_normalize('img', mean=[0.5, 0.4, 0.3],
                               std=[0.2, 0.25, 0.3])
    rng = np.random.RandomState(0)
    batch = rng.randint(0, 255, (4, 8, 6, 3)).astype(np.uint8)
    out = ts.func({'img': batch, 'label': np.arange(4)})
    expected = (batch.astype(np.float32) / 255.0 -
                np.array([0.5, 0.4, 0.3], np.float32)) / \
        np.array([0.2, 0.25, 0.3], np.float32)
    expected = expected.transpose(0, 3, 1, 2)
    np.testing.assert_allclose(out['img'], expected, rtol=1e-6)
    np.testing.assert_array_equal(out['label'], np.arange(4))
    # single row (row path)
    row = ts.func({'img': batch[0]})
    np.testing.assert_allclose(row['img'], expected[0], rtol=1e-6)
    # already-fused passthrough
    again = ts.func({'img': out['img']})
    assert again['img'] is out['img']


def test_fused_image_normalize_torch_path():
    import numpy as np
    import torch
    from petastorm_amd.transform import fused_image_normalize
    ts = fused_image_normalize('img', mean=[0.1, 0.2, 0.3],
                               std=[1.0, 2.0, 4.0])
    rng = np.random.RandomState(1)
    batch = torch.from_numpy(
        rng.randint(0, 255, (3, 5, 7, 3)).astype(np.uint8))
    out = ts.func({'img': batch})
    assert out['img'].shape == (3, 3, 5, 7)
    assert out['img'].dtype == torch.float32
    exp = (batch.float() / 255.0 -
           torch.tensor([0.1, 0.2, 0.3])) / torch.tensor([1.0, 2.0, 4.0])
    torch.testing.assert_close(out['img'], exp.permute(0, 3, 1, 2))
    assert ts.func({'img': out['img']})['img'] is out['img']
