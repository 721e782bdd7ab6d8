"""Synthetic dataset generators used by tests and benchmarks.

Parity role: /root/reference/petastorm/tests/test_common.py (the
``TestSchema`` / ``create_test_dataset`` fixtures :38-158) and the
HelloWorld dataset of the reference docs
(/root/reference/examples/hello_world/petastorm_dataset/generate_petastorm_dataset.py,
README.rst:78-92: id int32 + 128x256x3 png image + 4-D uint8 ndarray).

All data is random (no network access for real datasets).
"""

from decimal import Decimal

import numpy as np

from petastorm_amd.codecs import (CompressedImageCodec,
                                  CompressedNdarrayCodec, NdarrayCodec,
                                  ScalarCodec)
from petastorm_amd.etl.dataset_metadata import materialize_dataset
from petastorm_amd.unischema import Unischema, UnischemaField

# ---------------------------------------------------------------------------
# TestSchema analog: exercises every codec + nullable fields
# ---------------------------------------------------------------------------

TestSchema = Unischema('TestSchema', [
    UnischemaField('id', np.int64, (), ScalarCodec(), False),
    UnischemaField('id2', np.int32, (), ScalarCodec(), False),
    UnischemaField('id_float', np.float64, (), ScalarCodec(), False),
    UnischemaField('id_odd', np.bool_, (), ScalarCodec(), False),
    UnischemaField('python_primitive_uint8', np.uint8, (), ScalarCodec(), False),
    UnischemaField('image_png', np.uint8, (32, 16, 3),
                   CompressedImageCodec('png'), False),
    UnischemaField('matrix', np.float32, (10, 20), NdarrayCodec(), False),
    UnischemaField('decimal', Decimal, (), ScalarCodec(), False),
    UnischemaField('matrix_uint16', np.uint16, (2, 3), NdarrayCodec(), False),
    UnischemaField('matrix_uint32', np.uint32, (2, 3), NdarrayCodec(), False),
    UnischemaField('matrix_string', np.bytes_, (None,), NdarrayCodec(), False),
    UnischemaField('matrix_nullable', np.uint16, (2, 3), NdarrayCodec(), True),
    UnischemaField('sensor_name', np.str_, (), ScalarCodec(), False),
    UnischemaField('string_array_nullable', np.str_, (None,), NdarrayCodec(), True),
    UnischemaField('compressed_matrix', np.float32, (4, 5),
                   CompressedNdarrayCodec(), False),
])


def _random_test_row(rng, idx):
    return {
        'id': np.int64(idx),
        'id2': np.int32(idx % 10),
        'id_float': np.float64(idx),
        'id_odd': np.bool_(idx % 2),
        'python_primitive_uint8': np.uint8(rng.randint(0, 255)),
        'image_png': rng.randint(0, 255, (32, 16, 3)).astype(np.uint8),
        'matrix': rng.rand(10, 20).astype(np.float32),
        'decimal': Decimal(str(rng.randint(0, 1000)) + '.45'),
        'matrix_uint16': rng.randint(0, 2 ** 16, (2, 3)).astype(np.uint16),
        'matrix_uint32': rng.randint(0, 2 ** 16, (2, 3)).astype(np.uint32),
        'matrix_string': np.array([b'a' * (1 + rng.randint(3))
                                   for _ in range(rng.randint(1, 4))],
                                  dtype=np.bytes_),
        'matrix_nullable': (rng.randint(0, 2 ** 16, (2, 3)).astype(np.uint16)
                            if idx % 2 else None),
        'sensor_name': 'sensor-{}'.format(idx % 3),
        'string_array_nullable': (np.array(['abc', 'de'], dtype=np.str_)
                                  if idx % 3 else None),
        'compressed_matrix': rng.rand(4, 5).astype(np.float32),
    }


def create_test_dataset(url, num_rows=100, rowgroup_size_mb=1, seed=0,
                        num_files=2):
    """Write the TestSchema dataset; returns the list of source row dicts
    (pre-encode ground truth, like reference create_test_dataset)."""
    rng = np.random.RandomState(seed)
    rows = [_random_test_row(rng, i) for i in range(num_rows)]
    with materialize_dataset(url, TestSchema, rowgroup_size_mb) as writer:
        per_file = max(1, num_rows // num_files)
        for i, r in enumerate(rows):
            writer.write_row(r)
            if (i + 1) % per_file == 0 and i + 1 < num_rows:
                writer.new_file()
    return rows


# ---------------------------------------------------------------------------
# HelloWorld schema (the reference's headline benchmark dataset)
# ---------------------------------------------------------------------------

HelloWorldSchema = Unischema('HelloWorldSchema', [
    UnischemaField('id', np.int32, (), ScalarCodec(), False),
    UnischemaField('image1', np.uint8, (128, 256, 3),
                   CompressedImageCodec('png'), False),
    UnischemaField('array_4d', np.uint8, (None, 128, 30, None),
                   NdarrayCodec(), False),
])


def hello_world_row(rng, idx):
    """Same value shapes as the reference HelloWorld generator
    (examples/hello_world/petastorm_dataset/generate_petastorm_dataset.py)."""
    return {
        'id': np.int32(idx),
        'image1': rng.randint(0, 255, (128, 256, 3)).astype(np.uint8),
        'array_4d': rng.randint(0, 255,
                                (4, 128, 30, 3)).astype(np.uint8),
    }


def create_hello_world_dataset(url, num_rows=100, rowgroup_size_mb=16, seed=0):
    rng = np.random.RandomState(seed)
    with materialize_dataset(url, HelloWorldSchema, rowgroup_size_mb) as w:
        for i in range(num_rows):
            w.write_row(hello_world_row(rng, i))


# ---------------------------------------------------------------------------
# ImageNet-style schema (BASELINE config 3/4)
# ---------------------------------------------------------------------------

ImageNetSchema = Unischema('ImageNetSchema', [
    UnischemaField('image', np.uint8, (224, 224, 3),
                   CompressedImageCodec('jpeg', quality=90), False),
    UnischemaField('label', np.int32, (), ScalarCodec(), False),
])


def create_imagenet_dataset(url, num_rows=512, rowgroup_size_mb=32, seed=0,
                            structured=True, rows_per_rowgroup=None):
    """Random 224x224x3 jpegs + labels.

    ``structured`` images (smooth gradients + blobs) compress like photos;
    pure-noise images stress the Huffman decoder instead. Both are valid.
    """
    rng = np.random.RandomState(seed)
    yy, xx = np.mgrid[0:224, 0:224].astype(np.float32)
    # 16 reusable noise planes: per-image fresh noise costs ~3x the whole
    # generation (untimed, but it is driver wall-clock on fresh boxes)
    noise = [rng.randn(224, 224).astype(np.float32) * 10 for _ in range(16)]
    with materialize_dataset(url, ImageNetSchema, rowgroup_size_mb,
                             rows_per_rowgroup=rows_per_rowgroup) as w:
        for i in range(num_rows):
            if structured:
                base = (np.sin(xx / (8 + i % 13)) + np.cos(yy / (11 + i % 7)))
                img = np.stack([base * 60 + 128 + noise[(i + c) % 16]
                                for c in range(3)], axis=-1)
                img = np.clip(img, 0, 255).astype(np.uint8)
            else:
                img = rng.randint(0, 255, (224, 224, 3)).astype(np.uint8)
            w.write_row({'image': img, 'label': np.int32(i % 1000)})


# ---------------------------------------------------------------------------
# Scalar-only plain-parquet store (BASELINE config 2; reference
# tests/test_common.py:161-245 "scalar_dataset")
# ---------------------------------------------------------------------------

def create_scalar_dataset(url, num_rows=1000, num_float_cols=8,
                          num_int_cols=8, rowgroup_size=256, seed=0,
                          compression='snappy'):
    """Plain (non-petastorm) Parquet written directly through pyarrow."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    from petastorm_amd.fs_utils import get_filesystem_and_path_or_paths
    rng = np.random.RandomState(seed)
    cols = {'id': np.arange(num_rows, dtype=np.int64)}
    for i in range(num_float_cols):
        cols['f{}'.format(i)] = rng.rand(num_rows).astype(np.float64)
    for i in range(num_int_cols):
        cols['i{}'.format(i)] = rng.randint(0, 1 << 30, num_rows,
                                            dtype=np.int64)
    cols['name'] = np.array(['row-{}'.format(i) for i in range(num_rows)])
    table = pa.table(cols)
    fs, path = get_filesystem_and_path_or_paths(url)
    fs.makedirs(path, exist_ok=True)
    # 32 KiB pages: the page is the decompression-parallelism unit on the
    # GPU (one wave per snappy stream); small pages keep 256 CUs fed
    # (sweepable via PSA_SCALAR_PAGE_KB)
    import os
    page_kb = int(os.environ.get('PSA_SCALAR_PAGE_KB', '32'))
    pq.write_table(table, path + '/data-00000.parquet',
                   row_group_size=rowgroup_size, compression=compression,
                   use_dictionary=False, data_page_size=page_kb << 10)
    return cols


# ---------------------------------------------------------------------------
# NGram/sequence dataset (BASELINE config 5): 1024-token int32 rows
# ---------------------------------------------------------------------------

SequenceSchema = Unischema('SequenceSchema', [
    UnischemaField('timestamp', np.int64, (), ScalarCodec(), False),
    UnischemaField('tokens', np.int32, (1024,), NdarrayCodec(), False),
    UnischemaField('source', np.int32, (), ScalarCodec(), False),
])


def create_sequence_dataset(url, num_rows=200, rowgroup_size_mb=4, seed=0,
                            rows_per_rowgroup=None):
    rng = np.random.RandomState(seed)
    with materialize_dataset(url, SequenceSchema, rowgroup_size_mb,
                             rows_per_rowgroup=rows_per_rowgroup) as w:
        for i in range(num_rows):
            w.write_row({
                'timestamp': np.int64(i),
                'tokens': rng.randint(0, 50000, 1024).astype(np.int32),
                'source': np.int32(i % 4),
            })


def create_rich_scalar_dataset(url, num_rows=200, rowgroup_size=50):
    """Non-petastorm store with the reference's full scalar-type surface
    (reference tests/test_common.py:161-245): date, timestamp, strings,
    float64, fixed-size int list, nested struct."""
    import datetime
    import pyarrow as pa
    import pyarrow.parquet as pq
    from petastorm_amd.fs_utils import get_filesystem_and_path_or_paths
    rows = []
    for i in range(num_rows):
        rows.append({
            'id': np.int32(i),
            'id_div_700': np.int32(i // 700),
            'datetime': datetime.date(2019, 1, 2),
            'timestamp': datetime.datetime(2005, 2, 25, 3, 30),
            'string': 'hello_%d' % i,
            'string2': 'world_%d' % i,
            'float64': np.float64(i) * 0.66,
            'int_fixed_size_list': list(range(1 + i, 10 + i)),
            'nested_struct': {'nested_int': i},
        })
    table = pa.table({
        'id': pa.array([r['id'] for r in rows], pa.int32()),
        'id_div_700': pa.array([r['id_div_700'] for r in rows], pa.int32()),
        'datetime': pa.array([r['datetime'] for r in rows], pa.date32()),
        'timestamp': pa.array([r['timestamp'] for r in rows],
                              pa.timestamp('us')),
        'string': pa.array([r['string'] for r in rows], pa.string()),
        'string2': pa.array([r['string2'] for r in rows], pa.string()),
        'float64': pa.array([r['float64'] for r in rows], pa.float64()),
        'int_fixed_size_list': pa.array(
            [r['int_fixed_size_list'] for r in rows],
            pa.list_(pa.int32(), 9)),
        'nested_struct': pa.array(
            [r['nested_struct'] for r in rows],
            pa.struct([('nested_int', pa.int32())])),
    })
    fs, path = get_filesystem_and_path_or_paths(url)
    fs.makedirs(path, exist_ok=True)
    pq.write_table(table, path + '/data.parquet',
                   row_group_size=rowgroup_size, compression='none')
    return rows


def create_many_columns_dataset(url, num_rows=25, num_columns=1000):
    """1000-int32-column plain store (reference tests/test_common.py:
    248-294)."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    from petastorm_amd.fs_utils import get_filesystem_and_path_or_paths
    cols = {'col_{}'.format(c): np.arange(num_rows, dtype=np.int32) + c
            for c in range(num_columns)}
    table = pa.table(cols)
    fs, path = get_filesystem_and_path_or_paths(url)
    fs.makedirs(path, exist_ok=True)
    pq.write_table(table, path + '/data.parquet', row_group_size=10)
    return cols
