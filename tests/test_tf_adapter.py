"""Execute the TF adapter (tf_utils) against the faithful TF stand-in
(petastorm_amd.test_util.tf_stub): tf_tensors graph-mode reads, the
RandomShuffleQueue path, make_petastorm_dataset incl. auto-reset, and the
NGram flatten/unflatten shims — exercising the code paths the reference
covers in tests/test_tf_utils.py and tests/test_tf_dataset.py.
(TensorFlow itself is not installable offline; the stub models exactly
the API surface tf_utils touches — see tf_stub docstring.)"""
import sys
from decimal import Decimal

import numpy as np
import pytest

from petastorm_amd import make_batch_reader, make_reader
from petastorm_amd.ngram import NGram
from petastorm_amd.test_util import tf_stub
from petastorm_amd.test_util.dataset_gen import (TestSchema,
                                                 create_sequence_dataset,
                                                 create_test_dataset)


@pytest.fixture()
def tf(monkeypatch):
    mod = tf_stub.build_module()
    monkeypatch.setitem(sys.modules, 'tensorflow', mod)
    return mod


@pytest.fixture(scope='module')
def tf_dataset(tmp_path_factory):
    d = str(tmp_path_factory.mktemp('tfds'))
    url = 'file://' + d
    rows = create_test_dataset(url, num_rows=30, rowgroup_size_mb=1)
    return {'url': url, 'rows': rows}


def test_tf_tensors_reads_all_rows(tf, tf_dataset):
    from petastorm_amd.tf_utils import tf_tensors
    fields = ['id', 'id2', 'matrix', 'sensor_name', 'matrix_uint16',
              'decimal']
    expected = {r['id']: r for r in tf_dataset['rows']}
    with make_reader(tf_dataset['url'], schema_fields=fields,
                     reader_pool_type='dummy', num_epochs=1,
                     shuffle_row_groups=False) as reader:
        row_tensors = tf_tensors(reader)
        assert set(row_tensors._fields) == set(fields)
        with tf.compat.v1.Session() as sess:
            got = [sess.run(row_tensors) for _ in range(len(expected))]
    assert len(got) == len(expected)
    for row in got:
        src = expected[int(row.id)]
        np.testing.assert_allclose(row.matrix, src['matrix'], rtol=1e-6)
        # uint16 ndarray widened to int32 (reference tf_utils.py:57-96)
        assert row.matrix_uint16.dtype == np.int32
        np.testing.assert_array_equal(row.matrix_uint16,
                                      src['matrix_uint16'].astype(np.int32))
        # Decimal -> str
        assert isinstance(row.decimal, (str, bytes))
        assert Decimal(row.decimal if isinstance(row.decimal, str)
                       else row.decimal.decode()) == src['decimal']
        assert row.sensor_name == src['sensor_name']


def test_tf_tensors_static_shapes(tf, tf_dataset):
    from petastorm_amd.tf_utils import tf_tensors
    with make_reader(tf_dataset['url'], schema_fields=['id', 'matrix'],
                     reader_pool_type='dummy', num_epochs=1,
                     shuffle_row_groups=False) as reader:
        row_tensors = tf_tensors(reader)
        # fully-known shapes are set statically (reference :313-317)
        assert row_tensors.matrix.get_shape() == (10, 20)
        with tf.compat.v1.Session() as sess:
            row = sess.run(row_tensors)
    assert row.matrix.shape == (10, 20)


def test_tf_tensors_shuffling_queue(tf, tf_dataset):
    from petastorm_amd.tf_utils import tf_tensors
    n = 30
    with make_reader(tf_dataset['url'], schema_fields=['id'],
                     reader_pool_type='dummy', num_epochs=None,
                     shuffle_row_groups=False) as reader:
        row_tensors = tf_tensors(reader, shuffling_queue_capacity=20,
                                 min_after_dequeue=10)
        with tf.compat.v1.Session() as sess:
            ids = [int(sess.run(row_tensors).id) for _ in range(n)]
    # all values real rows; order perturbed by the queue
    assert set(ids) <= set(range(n * 2))
    assert len(ids) == n


def test_tf_tensors_batched_with_queue_rejected(tf, tf_dataset):
    from petastorm_amd.tf_utils import tf_tensors
    with make_batch_reader(tf_dataset['url'],
                           schema_fields=['id', 'id2'],
                           num_epochs=1) as reader:
        with pytest.raises(ValueError):
            tf_tensors(reader, shuffling_queue_capacity=10,
                       min_after_dequeue=5)


def test_tf_tensors_batched_reader(tf, tf_dataset):
    from petastorm_amd.tf_utils import tf_tensors
    with make_batch_reader(tf_dataset['url'], schema_fields=['id', 'id2'],
                           num_epochs=1, shuffle_row_groups=False) as reader:
        batch_tensors = tf_tensors(reader)
        with tf.compat.v1.Session() as sess:
            b = sess.run(batch_tensors)
    assert b.id.shape[0] > 0 and b.id.shape == b.id2.shape


def test_make_petastorm_dataset_and_auto_reset(tf, tf_dataset):
    from petastorm_amd.tf_utils import make_petastorm_dataset
    expected_ids = sorted(r['id'] for r in tf_dataset['rows'])
    with make_reader(tf_dataset['url'], schema_fields=['id', 'matrix'],
                     reader_pool_type='dummy', num_epochs=1,
                     shuffle_row_groups=False) as reader:
        ds = make_petastorm_dataset(reader)
        first = [int(row.id.numpy()) for row in ds]
        # re-iteration must auto reader.reset() (reference :374-380)
        second = [int(row.id.numpy()) for row in ds]
    assert sorted(first) == expected_ids
    assert sorted(second) == expected_ids


def test_make_petastorm_dataset_shape_validation(tf, tf_dataset):
    from petastorm_amd.tf_utils import make_petastorm_dataset
    with make_reader(tf_dataset['url'], schema_fields=['id', 'matrix'],
                     reader_pool_type='dummy', num_epochs=1,
                     shuffle_row_groups=False) as reader:
        ds = make_petastorm_dataset(reader)
        row = next(iter(ds))
        assert row.matrix.numpy().shape == (10, 20)


@pytest.fixture(scope='module')
def seq_url(tmp_path_factory):
    d = str(tmp_path_factory.mktemp('tfseq'))
    url = 'file://' + d
    create_sequence_dataset(url, num_rows=40, rowgroup_size_mb=1)
    return url


def _seq_ngram(length):
    return NGram(fields={i: ['timestamp', 'source'] for i in range(length)},
                 delta_threshold=10 ** 9, timestamp_field='timestamp')


def test_tf_tensors_ngram(tf, seq_url):
    from petastorm_amd.tf_utils import tf_tensors
    ng = _seq_ngram(3)
    with make_reader(seq_url, schema_fields=ng, reader_pool_type='dummy',
                     num_epochs=1, shuffle_row_groups=False) as reader:
        ngram_tensors = tf_tensors(reader)
        assert sorted(ngram_tensors.keys()) == [0, 1, 2]
        with tf.compat.v1.Session() as sess:
            got = sess.run(ngram_tensors)
    # timestamps strictly increasing across timesteps of the window
    ts = [int(got[i].timestamp) for i in range(3)]
    assert ts[0] < ts[1] < ts[2]


def test_make_petastorm_dataset_ngram(tf, seq_url):
    from petastorm_amd.tf_utils import make_petastorm_dataset
    ng = _seq_ngram(2)
    with make_reader(seq_url, schema_fields=ng, reader_pool_type='dummy',
                     num_epochs=1, shuffle_row_groups=False) as reader:
        ds = make_petastorm_dataset(reader)
        windows = list(ds)
    assert windows
    for w in windows:
        assert sorted(w.keys()) == [0, 1]
        assert int(w[0].timestamp.numpy()) < int(w[1].timestamp.numpy())
