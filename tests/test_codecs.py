"""Codec round-trips (parity: reference tests/test_codec_*.py)."""
from decimal import Decimal

import numpy as np
import pytest

from petastorm_amd.codecs import (CompressedImageCodec,
                                  CompressedNdarrayCodec, NdarrayCodec,
                                  ScalarCodec, codec_from_dict, codec_to_dict)
from petastorm_amd.unischema import UnischemaField


def _field(name, dtype, shape, codec, nullable=False):
    return UnischemaField(name, dtype, shape, codec, nullable)


def test_png_roundtrip_lossless_rgb():
    codec = CompressedImageCodec('png')
    f = _field('im', np.uint8, (10, 12, 3), codec)
    img = np.random.RandomState(0).randint(0, 255, (10, 12, 3)).astype(np.uint8)
    data = codec.encode(f, img)
    assert isinstance(data, bytes) and data[:8] == b'\x89PNG\r\n\x1a\n'
    out = codec.decode(f, data)
    np.testing.assert_array_equal(out, img)


def test_png_roundtrip_grayscale_uint16():
    codec = CompressedImageCodec('png')
    f = _field('im', np.uint16, (6, 8), codec)
    img = np.random.RandomState(0).randint(0, 2 ** 16, (6, 8)).astype(np.uint16)
    out = codec.decode(f, codec.encode(f, img))
    np.testing.assert_array_equal(out, img)


def test_jpeg_roundtrip_lossy_close():
    codec = CompressedImageCodec('jpeg', quality=95)
    f = _field('im', np.uint8, (32, 32, 3), codec)
    # smooth image so jpeg error is small (reference
    # tests/test_codec_compressed_image.py uses the same trick)
    y, x = np.mgrid[0:32, 0:32]
    img = np.stack([(x * 4) % 255, (y * 4) % 255, ((x + y) * 2) % 255],
                   axis=-1).astype(np.uint8)
    out = codec.decode(f, codec.encode(f, img))
    assert out.shape == img.shape
    assert np.abs(out.astype(int) - img.astype(int)).mean() < 10


def test_jpeg_has_restart_markers():
    codec = CompressedImageCodec('jpeg', quality=90)
    f = _field('im', np.uint8, (64, 64, 3), codec)
    img = np.random.RandomState(0).randint(0, 255, (64, 64, 3)).astype(np.uint8)
    data = codec.encode(f, img)
    n_rst = sum(1 for i in range(len(data) - 1)
                if data[i] == 0xFF and 0xD0 <= data[i + 1] <= 0xD7)
    assert n_rst >= 1  # one per MCU row - the GPU parallel-decode hook


def test_ndarray_roundtrip():
    codec = NdarrayCodec()
    f = _field('m', np.float32, (5, 6), codec)
    m = np.random.rand(5, 6).astype(np.float32)
    data = codec.encode(f, m)
    assert data[:6] == b'\x93NUMPY'
    np.testing.assert_array_equal(codec.decode(f, data), m)


def test_ndarray_wrong_dtype_raises():
    codec = NdarrayCodec()
    f = _field('m', np.float32, (5, 6), codec)
    with pytest.raises(ValueError):
        codec.encode(f, np.zeros((5, 6), np.float64))


def test_ndarray_wrong_shape_raises():
    codec = NdarrayCodec()
    f = _field('m', np.float32, (5, 6), codec)
    with pytest.raises(ValueError):
        codec.encode(f, np.zeros((5, 7), np.float32))


def test_ndarray_none_dims_match_anything():
    codec = NdarrayCodec()
    f = _field('m', np.float32, (None, 6), codec)
    m = np.zeros((9, 6), np.float32)
    np.testing.assert_array_equal(codec.decode(f, codec.encode(f, m)), m)


def test_compressed_ndarray_roundtrip():
    codec = CompressedNdarrayCodec()
    f = _field('m', np.int32, (100,), codec)
    m = np.arange(100, dtype=np.int32)
    data = codec.encode(f, m)
    assert len(data) < m.nbytes  # actually compressed
    np.testing.assert_array_equal(codec.decode(f, data), m)


def test_scalar_codec_types():
    codec = ScalarCodec()
    cases = [
        (np.int32, 42, np.int32(42)),
        (np.float64, 2.5, np.float64(2.5)),
        (np.str_, 'hello', 'hello'),
        (Decimal, Decimal('1.23'), Decimal('1.23')),
        (np.bool_, True, np.bool_(True)),
    ]
    for dtype, value, expected in cases:
        f = _field('s', dtype, (), codec)
        out = codec.decode(f, codec.encode(f, value))
        assert out == expected, (dtype, out)


def test_codec_json_roundtrip():
    for codec in [ScalarCodec(), NdarrayCodec(), CompressedNdarrayCodec(9),
                  CompressedImageCodec('jpeg', 75)]:
        restored = codec_from_dict(codec_to_dict(codec))
        assert type(restored) is type(codec)
    assert codec_from_dict(None) is None
    assert codec_from_dict(codec_to_dict(
        CompressedImageCodec('jpeg', 75))).quality == 75


def test_compressed_ndarray_npz_container_upstream_compatible():
    """container='npz' writes np.savez_compressed payloads byte-compatible
    with upstream petastorm's CompressedNdarrayCodec; schema JSON keeps
    the container through a round trip."""
    import io
    import numpy as np
    from petastorm_amd.codecs import (CompressedNdarrayCodec, codec_from_dict)
    from petastorm_amd.unischema import UnischemaField
    arr = np.arange(24, dtype=np.int32).reshape(4, 6)
    field = UnischemaField('m', np.int32, (4, 6),
                           CompressedNdarrayCodec(container='npz'), False)
    blob = field.codec.encode(field, arr)
    assert blob[:2] == b'PK'  # npz (zip) container, as upstream writes
    with np.load(io.BytesIO(blob), allow_pickle=False) as npz:
        np.testing.assert_array_equal(npz[npz.files[0]], arr)
    np.testing.assert_array_equal(field.codec.decode(field, blob), arr)
    back = codec_from_dict(field.codec.to_dict())
    assert back.container == 'npz'
    np.testing.assert_array_equal(back.decode(field, back.encode(field, arr)),
                                  arr)


def test_image_codec_single_channel_shape_roundtrip():
    """(H, W, 1) image fields come back with the DECLARED shape even though
    the container stores a 2-D grayscale image."""
    from petastorm_amd.codecs import CompressedImageCodec
    from petastorm_amd.unischema import UnischemaField
    rng = np.random.RandomState(0)
    for codec_name in ('png', 'jpeg'):
        f = UnischemaField('g', np.uint8, (28, 28, 1),
                           CompressedImageCodec(codec_name, quality=95),
                           False)
        v = rng.randint(0, 255, (28, 28, 1)).astype(np.uint8)
        out = f.codec.decode(f, f.codec.encode(f, v))
        assert out.shape == (28, 28, 1)
        if codec_name == 'png':
            np.testing.assert_array_equal(out, v)


def test_float16_scalar_roundtrip_both_routes(tmp_path):
    """float16 scalars (stored as Parquet FLBA/half) survive write->read
    exactly on the row and batch routes."""
    from petastorm_amd import make_batch_reader, make_reader
    from petastorm_amd.codecs import ScalarCodec
    from petastorm_amd.etl.dataset_metadata import materialize_dataset
    from petastorm_amd.unischema import Unischema, UnischemaField
    S = Unischema('F16', [
        UnischemaField('id', np.int64, (), ScalarCodec(), False),
        UnischemaField('h', np.float16, (), ScalarCodec(), False),
    ])
    url = 'file://' + str(tmp_path / 'f16')
    rng = np.random.RandomState(0)
    vals = rng.rand(50).astype(np.float16)
    with materialize_dataset(url, S, rowgroup_size_mb=1) as w:
        for i in range(50):
            w.write_row({'id': np.int64(i), 'h': vals[i]})
    with make_reader(url, reader_pool_type='dummy',
                     shuffle_row_groups=False) as r:
        got = {int(row.id): row.h for row in r}
    assert all(got[i] == vals[i] for i in range(50))
    with make_batch_reader(url, shuffle_row_groups=False) as r:
        b = next(iter(r))
    assert b.h.dtype == np.float16
    np.testing.assert_array_equal(np.asarray(b.h),
                                  vals[np.asarray(b.id)])
