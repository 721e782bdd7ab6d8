"""Build a dataset laid out exactly like UPSTREAM petastorm writes it:
plain-parquet data files plus a ``_common_metadata`` file whose key-value
metadata carries a *pickled* ``petastorm.unischema.Unischema``
(reference etl/dataset_metadata.py:194-205) — petastorm itself is NOT
installed here, so the pickle stream is produced through stand-in modules
registered under the reference's import paths for the duration of the
write.  Used by the interop tests and available for ad-hoc fixture
generation.
"""

import contextlib
import io
import pickle
import sys
import types
from collections import OrderedDict, namedtuple

import numpy as np


@contextlib.contextmanager
def fake_reference_modules():
    """Register minimal petastorm/pyspark modules so ``pickle.dumps`` emits
    the same GLOBAL paths a real reference store contains.  The class
    *layouts* mirror the reference (attribute names are the pickle
    contract): Unischema(_name, _fields + per-field attrs)
    (reference unischema.py:179-196), UnischemaField 5-slot namedtuple
    (unischema.py:50-68), codec attrs (codecs.py:67-68,223)."""
    created = {}

    def module(name):
        m = types.ModuleType(name)
        created[name] = m
        return m

    ps = module('petastorm')
    ps_uni = module('petastorm.unischema')
    ps_codecs = module('petastorm.codecs')
    pyspark = module('pyspark')
    pyspark_sql = module('pyspark.sql')
    pyspark_types = module('pyspark.sql.types')
    ps.unischema = ps_uni
    ps.codecs = ps_codecs
    pyspark.sql = pyspark_sql
    pyspark_sql.types = pyspark_types

    UnischemaField = namedtuple(
        'UnischemaField', ['name', 'numpy_dtype', 'shape', 'codec',
                           'nullable'])
    UnischemaField.__module__ = 'petastorm.unischema'
    ps_uni.UnischemaField = UnischemaField

    class Unischema(object):
        def __init__(self, name, fields):
            self._name = name
            fields = sorted(fields, key=lambda t: t.name)
            self._fields = OrderedDict([(f.name, f) for f in fields])
            for f in fields:
                setattr(self, f.name, f)
    Unischema.__module__ = 'petastorm.unischema'
    Unischema.__qualname__ = 'Unischema'
    ps_uni.Unischema = Unischema

    class CompressedImageCodec(object):
        def __init__(self, image_codec='png', quality=80):
            self._image_codec = '.' + image_codec
            self._quality = quality
    CompressedImageCodec.__module__ = 'petastorm.codecs'
    CompressedImageCodec.__qualname__ = 'CompressedImageCodec'
    ps_codecs.CompressedImageCodec = CompressedImageCodec

    class NdarrayCodec(object):
        pass
    NdarrayCodec.__module__ = 'petastorm.codecs'
    NdarrayCodec.__qualname__ = 'NdarrayCodec'
    ps_codecs.NdarrayCodec = NdarrayCodec

    class CompressedNdarrayCodec(object):
        pass
    CompressedNdarrayCodec.__module__ = 'petastorm.codecs'
    CompressedNdarrayCodec.__qualname__ = 'CompressedNdarrayCodec'
    ps_codecs.CompressedNdarrayCodec = CompressedNdarrayCodec

    class ScalarCodec(object):
        def __init__(self, spark_type):
            self._spark_type = spark_type
    ScalarCodec.__module__ = 'petastorm.codecs'
    ScalarCodec.__qualname__ = 'ScalarCodec'
    ps_codecs.ScalarCodec = ScalarCodec

    for tname in ('StringType', 'IntegerType', 'LongType', 'ShortType',
                  'ByteType', 'FloatType', 'DoubleType', 'BooleanType'):
        cls = type(tname, (object,), {})
        cls.__module__ = 'pyspark.sql.types'
        setattr(pyspark_types, tname, cls)

    class DecimalType(object):
        def __init__(self, precision=10, scale=0):
            self.precision = precision
            self.scale = scale
    DecimalType.__module__ = 'pyspark.sql.types'
    DecimalType.__qualname__ = 'DecimalType'
    pyspark_types.DecimalType = DecimalType

    saved = {k: sys.modules.get(k) for k in created}
    sys.modules.update(created)
    try:
        yield types.SimpleNamespace(
            Unischema=Unischema, UnischemaField=UnischemaField,
            CompressedImageCodec=CompressedImageCodec,
            NdarrayCodec=NdarrayCodec,
            CompressedNdarrayCodec=CompressedNdarrayCodec,
            ScalarCodec=ScalarCodec, spark_types=pyspark_types)
    finally:
        for k, v in saved.items():
            if v is None:
                sys.modules.pop(k, None)
            else:
                sys.modules[k] = v


def _encode_png(arr):
    from PIL import Image
    buf = io.BytesIO()
    Image.fromarray(arr).save(buf, format='PNG')
    return buf.getvalue()


def _encode_npy(arr):
    buf = io.BytesIO()
    np.save(buf, arr)
    return buf.getvalue()


def _encode_npz(arr):
    buf = io.BytesIO()
    np.savez_compressed(buf, arr)
    return buf.getvalue()


def create_reference_style_dataset(path, num_rows=20, rows_per_group=5,
                                   seed=0):
    """Write <path>/ as an upstream-petastorm store: parquet data files +
    _common_metadata with the pickled Unischema.  Returns the list of
    expected decoded row dicts."""
    import pyarrow as pa
    import pyarrow.parquet as pq

    rng = np.random.RandomState(seed)
    rows = []
    for i in range(num_rows):
        rows.append({
            'id': np.int32(i),
            'image_png': rng.randint(0, 255, (16, 24, 3)).astype(np.uint8),
            'embedding': rng.randn(8).astype(np.float32),
            'matrix_z': rng.randint(0, 9, (4, 5)).astype(np.int64),
            'label': 'item_%d' % i,
        })

    with fake_reference_modules() as ref:
        t = ref.spark_types
        schema = ref.Unischema('InteropSchema', [
            ref.UnischemaField('id', np.int32, (),
                               ref.ScalarCodec(t.IntegerType()), False),
            ref.UnischemaField('image_png', np.uint8, (16, 24, 3),
                               ref.CompressedImageCodec('png'), False),
            ref.UnischemaField('embedding', np.float32, (8,),
                               ref.NdarrayCodec(), False),
            ref.UnischemaField('matrix_z', np.int64, (4, 5),
                               ref.CompressedNdarrayCodec(), False),
            ref.UnischemaField('label', np.str_, (),
                               ref.ScalarCodec(t.StringType()), False),
        ])
        pickled_schema = pickle.dumps(schema, protocol=2)

    encoded = {
        'id': [int(r['id']) for r in rows],
        'image_png': [_encode_png(r['image_png']) for r in rows],
        'embedding': [_encode_npy(r['embedding']) for r in rows],
        'matrix_z': [_encode_npz(r['matrix_z']) for r in rows],
        'label': [r['label'] for r in rows],
    }
    arrow_schema = pa.schema([
        pa.field('id', pa.int32(), nullable=False),
        pa.field('image_png', pa.binary(), nullable=False),
        pa.field('embedding', pa.binary(), nullable=False),
        pa.field('matrix_z', pa.binary(), nullable=False),
        pa.field('label', pa.string(), nullable=False),
    ])
    table = pa.Table.from_pydict(encoded, schema=arrow_schema)

    import os
    os.makedirs(path, exist_ok=True)
    pq.write_table(table, os.path.join(path, 'part-00000.parquet'),
                   row_group_size=rows_per_group, use_dictionary=False)

    meta_schema = arrow_schema.with_metadata({
        b'dataset-toolkit.unischema.v1': pickled_schema,
        b'dataset-toolkit.num_row_groups.v1':
            ('{"part-00000.parquet": %d}'
             % ((num_rows + rows_per_group - 1) // rows_per_group)).encode(),
    })
    pq.write_metadata(meta_schema, os.path.join(path, '_common_metadata'))
    return rows
