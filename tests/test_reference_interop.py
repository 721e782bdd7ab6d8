"""Interop with datasets written by UPSTREAM petastorm: the pickled
Unischema in _common_metadata must load through the restricted depickler
and codec fields must decode to their original values
(reference etl/dataset_metadata.py:194-205,356-385; etl/legacy.py:22-79;
VERDICT r1 missing item 3)."""
import pickle

import numpy as np
import pytest

from petastorm_amd import make_reader
from petastorm_amd.etl import interop
from petastorm_amd.etl.dataset_metadata import get_schema
from petastorm_amd.fs_utils import get_filesystem_and_path_or_paths
from petastorm_amd.test_util.reference_store import (
    create_reference_style_dataset, fake_reference_modules)


@pytest.fixture(scope='module')
def reference_store(tmp_path_factory):
    d = str(tmp_path_factory.mktemp('ref_store'))
    rows = create_reference_style_dataset(d, num_rows=20, rows_per_group=5)
    return {'path': d, 'url': 'file://' + d, 'rows': rows}


def test_get_schema_loads_pickled_reference_metadata(reference_store):
    fs, path = get_filesystem_and_path_or_paths(reference_store['url'])
    schema = get_schema(fs, path)
    assert set(schema.fields) == {'id', 'image_png', 'embedding',
                                  'matrix_z', 'label'}
    f = schema.fields['image_png']
    assert f.shape == (16, 24, 3) and f.numpy_dtype == np.uint8
    assert type(f.codec).__name__ == 'CompressedImageCodec'
    assert f.codec.image_codec == 'png'
    assert type(schema.fields['embedding'].codec).__name__ == 'NdarrayCodec'
    assert type(schema.fields['matrix_z'].codec).__name__ == \
        'CompressedNdarrayCodec'
    assert type(schema.fields['id'].codec).__name__ == 'ScalarCodec'


def test_make_reader_decodes_reference_store(reference_store):
    got = {}
    with make_reader(reference_store['url'],
                     shuffle_row_groups=False) as reader:
        for row in reader:
            got[int(row.id)] = row
    assert len(got) == len(reference_store['rows'])
    for expected in reference_store['rows']:
        row = got[int(expected['id'])]
        np.testing.assert_array_equal(row.image_png, expected['image_png'])
        np.testing.assert_array_equal(row.embedding, expected['embedding'])
        np.testing.assert_array_equal(row.matrix_z, expected['matrix_z'])
        assert row.label == expected['label']


def test_rowgroup_counts_sidecar(reference_store):
    fs, path = get_filesystem_and_path_or_paths(reference_store['url'])
    counts = interop.load_reference_rowgroup_counts(fs, path)
    assert counts == {'part-00000.parquet': 4}


def test_forbidden_global_rejected():
    blob = pickle.dumps(pickle.Unpickler)  # class from module 'pickle'
    with pytest.raises(pickle.UnpicklingError):
        interop.restricted_loads(blob)


def test_unknown_petastorm_symbol_rejected():
    # A petastorm.* symbol outside the supported schema surface must not
    # silently resolve.
    blob = (b'cpetastorm.utils\nrun_in_subprocess\n.')
    with pytest.raises(pickle.UnpicklingError):
        interop.restricted_loads(blob)


def test_legacy_package_rename():
    """Pickles from pre-rename petastorm ('av.ml.dataset_toolkit.*') must
    load after the stream-level rename (reference etl/legacy.py:54-79)."""
    with fake_reference_modules() as ref:
        schema = ref.Unischema('Legacy', [
            ref.UnischemaField('x', np.int32, (), None, False)])
        blob = pickle.dumps(schema, protocol=0)
    legacy_blob = blob.replace(b'\ncpetastorm.unischema\n',
                               b'\ncav.ml.dataset_toolkit.unischema\n')
    # protocol 0 GLOBAL lines: "c<module>\n<name>\n" preceded by '(' marks;
    # replicate the reference's byte-level match window.
    legacy_blob = blob.replace(b'(cpetastorm.unischema\n',
                               b'(cav.ml.dataset_toolkit.unischema\n')
    assert legacy_blob != blob
    shim = interop.restricted_loads(
        interop._apply_legacy_renames(legacy_blob))
    out = interop.convert_reference_unischema(shim)
    assert list(out.fields) == ['x']


def test_npz_payload_decodes_via_compressed_ndarray_codec():
    """Upstream CompressedNdarrayCodec writes np.savez_compressed
    containers (reference codecs.py:193-198); decode must accept them."""
    import io
    from petastorm_amd.codecs import CompressedNdarrayCodec
    from petastorm_amd.unischema import UnischemaField
    arr = np.arange(12, dtype=np.float64).reshape(3, 4)
    buf = io.BytesIO()
    np.savez_compressed(buf, arr)
    field = UnischemaField('m', np.float64, (3, 4),
                           CompressedNdarrayCodec(), False)
    out = field.codec.decode(field, buf.getvalue())
    np.testing.assert_array_equal(out, arr)
    # and our own zlib-npy framing still round-trips
    own = field.codec.encode(field, arr)
    np.testing.assert_array_equal(field.codec.decode(field, own), arr)


@pytest.mark.gpu
def test_gpu_batch_reader_on_reference_store(tmp_path):
    """The GPU route must decode an upstream-petastorm-written store
    (pickled metadata, cv2-era png/npy/npz payloads): png + ndarray
    columns on device, values matching the originals."""
    import torch
    assert torch.cuda.is_available()
    from petastorm_amd import make_batch_reader
    d = str(tmp_path / 'ref_gpu')
    rows = create_reference_style_dataset(d, num_rows=20, rows_per_group=5)
    with make_batch_reader('file://' + d, device='cuda',
                           shuffle_row_groups=False) as r:
        batches = list(r)
        assist = r.diagnostics['cpu_assist_columns']
    assert 'image_png' not in assist and 'id' not in assist
    got_img, got_emb, got_mat = {}, {}, {}
    for b in batches:
        ids = b.id.cpu().numpy()
        for i, rid in enumerate(ids):
            got_img[int(rid)] = b.image_png[i].cpu().numpy()
            e = b.embedding[i]
            got_emb[int(rid)] = e.cpu().numpy() if hasattr(e, 'cpu') else e
            m = b.matrix_z[i]
            got_mat[int(rid)] = m.cpu().numpy() if hasattr(m, 'cpu') else m
    assert len(got_img) == len(rows)
    for src in rows:
        rid = int(src['id'])
        np.testing.assert_array_equal(got_img[rid], src['image_png'])
        np.testing.assert_array_equal(got_emb[rid], src['embedding'])
        np.testing.assert_array_equal(got_mat[rid], src['matrix_z'])

def test_dangerous_builtins_rejected():
    """builtins/os symbols must not resolve even though a few inert
    builtins (object, set, ...) are allow-listed by name."""
    for blob in (b'cbuiltins\neval\n.', b'c__builtin__\nexec\n.',
                 b'cos\nsystem\n.', b'cbuiltins\ngetattr\n.',
                 b'cnumpy.testing\nassert_equal\n.'):
        with pytest.raises(pickle.UnpicklingError):
            interop.restricted_loads(blob)
    # the allow-listed inert ones still work
    assert interop.restricted_loads(pickle.dumps({1, 2})) == {1, 2}
    import numpy as _np
    assert interop.restricted_loads(pickle.dumps(_np.int32)) is _np.int32
    assert interop.restricted_loads(
        pickle.dumps(_np.dtype('float32'))) == _np.dtype('float32')

def test_numpy1_legacy_scalar_names():
    """Stores pickled under numpy 1.x reference removed aliases
    (numpy.unicode_ / numpy.string_); they must map to modern classes."""
    import numpy as _np
    assert interop.restricted_loads(b'cnumpy\nunicode_\n.') is _np.str_
    assert interop.restricted_loads(b'cnumpy\nstring_\n.') is _np.bytes_
    assert interop.restricted_loads(b'cnumpy\nfloat_\n.') is _np.float64
