"""Unischema behavior (parity with reference tests/test_unischema.py)."""
from decimal import Decimal

import numpy as np
import pytest

from petastorm_amd.codecs import NdarrayCodec, ScalarCodec
from petastorm_amd.unischema import (Unischema, UnischemaField,
                                     dict_to_encoded_row,
                                     insert_explicit_nulls,
                                     match_unischema_fields)

SampleSchema = Unischema('SampleSchema', [
    UnischemaField('id', np.int64, (), ScalarCodec(), False),
    UnischemaField('value', np.float32, (), ScalarCodec(), True),
    UnischemaField('mat', np.float32, (3, 4), NdarrayCodec(), False),
    UnischemaField('other_mat', np.uint8, (2, 2), NdarrayCodec(), False),
])


def test_fields_ordered_and_attribute_access():
    assert list(SampleSchema.fields.keys()) == sorted(
        ['id', 'value', 'mat', 'other_mat'])
    assert SampleSchema.id.numpy_dtype is np.int64
    with pytest.raises(AttributeError):
        SampleSchema.nonexistent


def test_field_equality_and_hash():
    a = UnischemaField('x', np.int32, (), ScalarCodec(), False)
    b = UnischemaField('x', np.int32, (), ScalarCodec(), False)
    c = UnischemaField('x', np.int64, (), ScalarCodec(), False)
    assert a == b and hash(a) == hash(b)
    assert a != c


def test_create_schema_view_by_field():
    view = SampleSchema.create_schema_view([SampleSchema.id])
    assert list(view.fields.keys()) == ['id']


def test_create_schema_view_by_regex():
    view = SampleSchema.create_schema_view(['.*mat'])
    assert set(view.fields.keys()) == {'mat', 'other_mat'}
    # regex is a full match: 'mat' alone matches only 'mat'
    view2 = SampleSchema.create_schema_view(['mat'])
    assert set(view2.fields.keys()) == {'mat'}


def test_create_schema_view_foreign_field_raises():
    foreign = UnischemaField('foreign', np.int32, (), ScalarCodec(), False)
    with pytest.raises(ValueError):
        SampleSchema.create_schema_view([foreign])


def test_match_unischema_fields_mixed():
    got = match_unischema_fields(SampleSchema, ['id', SampleSchema.mat])
    assert {f.name for f in got} == {'id', 'mat'}


def test_namedtuple_caching():
    t1 = SampleSchema._get_namedtuple()
    t2 = SampleSchema._get_namedtuple()
    assert t1 is t2
    row = SampleSchema.make_namedtuple(id=1, value=2.0,
                                       mat=np.zeros((3, 4), np.float32),
                                       other_mat=np.zeros((2, 2), np.uint8))
    assert row.id == 1


def test_json_roundtrip():
    s = Unischema.from_json(SampleSchema.to_json())
    assert list(s.fields.keys()) == list(SampleSchema.fields.keys())
    for name in s.fields:
        assert s.fields[name] == SampleSchema.fields[name]


def test_insert_explicit_nulls():
    row = {'id': 1, 'mat': np.zeros((3, 4), np.float32),
           'other_mat': np.zeros((2, 2), np.uint8)}
    insert_explicit_nulls(SampleSchema, row)
    assert row['value'] is None
    with pytest.raises(ValueError):
        insert_explicit_nulls(SampleSchema, {'value': 1.0})


def test_dict_to_encoded_row_key_mismatch():
    with pytest.raises(ValueError):
        dict_to_encoded_row(SampleSchema, {'id': 1, 'bogus': 2})


def test_from_arrow_schema():
    import pyarrow as pa
    arrow = pa.schema([
        pa.field('a', pa.int32()),
        pa.field('b', pa.string()),
        pa.field('c', pa.list_(pa.float32())),
        pa.field('d', pa.decimal128(10, 2)),
    ])
    s = Unischema.from_arrow_schema(arrow)
    assert s.a.numpy_dtype is np.int32 and s.a.shape == ()
    assert s.b.numpy_dtype is np.str_
    assert s.c.numpy_dtype is np.float32 and s.c.shape == (None,)
    assert s.d.numpy_dtype is Decimal


def test_from_arrow_schema_unsupported_warns():
    import pyarrow as pa
    arrow = pa.schema([
        pa.field('ok', pa.int64()),
        pa.field('bad', pa.struct([pa.field('x', pa.int32())])),
    ])
    with pytest.warns(UserWarning):
        s = Unischema.from_arrow_schema(arrow)
    assert list(s.fields.keys()) == ['ok']


def test_many_fields_namedtuple():
    # >255 fields must work on modern python (reference needed a workaround,
    # namedtuple_gt_255_fields.py)
    fields = [UnischemaField('f{:04d}'.format(i), np.int32, (), None, False)
              for i in range(300)]
    s = Unischema('big', fields)
    nt = s.make_namedtuple(**{'f{:04d}'.format(i): i for i in range(300)})
    assert nt.f0299 == 299


def test_namedtuple_cache_shared_across_instances():
    """Two Unischema instances with the same name+fields share one
    namedtuple class, so rows from different readers compare equal
    (reference unischema.py:88-111 _NamedtupleCache)."""
    from petastorm_amd.codecs import ScalarCodec
    from petastorm_amd.unischema import Unischema, UnischemaField

    def build():
        return Unischema('CacheT', [
            UnischemaField('a', np.int32, (), ScalarCodec(), False),
            UnischemaField('b', np.float32, (), ScalarCodec(), False)])

    s1, s2 = build(), build()
    r1 = s1.make_namedtuple(a=1, b=2.0)
    r2 = s2.make_namedtuple(a=1, b=2.0)
    assert type(r1) is type(r2)
    assert r1 == r2
