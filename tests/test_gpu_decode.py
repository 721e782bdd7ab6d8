"""GPU kernel numerics tests: every HIP kernel vs a plain CPU reference.

All tests in this file require an MI355X (marked gpu).
"""
import io

import numpy as np
import pytest
import torch

from petastorm_amd import ops

pytestmark = pytest.mark.gpu


@pytest.fixture(scope='module')
def ext():
    assert torch.cuda.is_available(), 'gpu tests need a GPU'
    return ops.ext()


# ---------------------------------------------------------------------------
# snappy
# ---------------------------------------------------------------------------

def _snappy_compress(data):
    import pyarrow as pa
    return pa.compress(data, codec='snappy', asbytes=True)


def test_snappy_roundtrip_random_and_repetitive(ext):
    rng = np.random.RandomState(0)
    payloads = [
        rng.randint(0, 255, 100000).astype(np.uint8).tobytes(),   # literals
        (b'abcdefgh' * 20000),                                    # long copies
        (b'\x00' * 65536),                                        # RLE-ish
        rng.randint(0, 4, 50000).astype(np.uint8).tobytes(),      # mixed
        b'x',                                                     # tiny
    ]
    comp = [_snappy_compress(p) for p in payloads]
    comp_cat = b''.join(comp)
    c_off = np.zeros(len(comp) + 1, dtype=np.int64)
    c_off[1:] = np.cumsum([len(c) for c in comp])
    u_off = np.zeros(len(payloads) + 1, dtype=np.int64)
    u_off[1:] = np.cumsum([len(p) for p in payloads])

    dev = 'cuda'
    comp_t = torch.frombuffer(bytearray(comp_cat + b'\0' * 16),
                              dtype=torch.uint8).to(dev)
    out = torch.zeros(int(u_off[-1]) + 16, dtype=torch.uint8, device=dev)
    status = torch.zeros(len(payloads), dtype=torch.int32, device=dev)
    ext.snappy_decompress_batch(comp_t,
                                torch.from_numpy(c_off[:-1]).to(dev),
                                torch.from_numpy(c_off[1:]).to(dev),
                                out, torch.from_numpy(u_off[:-1]).to(dev),
                                torch.from_numpy(np.diff(u_off)).to(dev),
                                status)
    torch.cuda.synchronize()
    assert status.cpu().tolist() == [0] * len(payloads)
    got = out[:int(u_off[-1])].cpu().numpy().tobytes()
    assert got == b''.join(payloads)


# ---------------------------------------------------------------------------
# RLE/bit-packed hybrid
# ---------------------------------------------------------------------------

def _write_uvarint(v):
    out = bytearray()
    while True:
        b = v & 0x7f
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _encode_hybrid(values, bit_width):
    """Simple encoder: alternating RLE and bit-packed runs."""
    out = bytearray()
    i = 0
    n = len(values)
    while i < n:
        # run-length detect
        j = i
        while j < n and values[j] == values[i]:
            j += 1
        if j - i >= 8:
            out += _write_uvarint((j - i) << 1)
            v = int(values[i])
            for k in range((bit_width + 7) // 8):
                out.append((v >> (8 * k)) & 0xFF)
            i = j
        else:
            # bit-pack the next up-to-504 values in groups of 8
            take = min(n - i, 504)
            groups = (take + 7) // 8
            out += _write_uvarint((groups << 1) | 1)
            bits = 0
            acc = 0
            cnt = 0
            for k in range(groups * 8):
                v = int(values[i + k]) if i + k < n else 0
                acc |= v << bits
                bits += bit_width
                while bits >= 8:
                    out.append(acc & 0xFF)
                    acc >>= 8
                    bits -= 8
                cnt += 1
            if bits:
                out.append(acc & 0xFF)
            i += take
    return bytes(out)


@pytest.mark.parametrize('bit_width', [1, 2, 5, 8, 12, 20])
def test_rle_hybrid_decode(ext, bit_width):
    rng = np.random.RandomState(bit_width)
    maxv = (1 << bit_width) - 1
    streams, expected = [], []
    for s in range(6):
        n = rng.randint(1, 3000)
        vals = rng.randint(0, maxv + 1, n)
        if s % 2 == 0:
            vals[: n // 2] = vals[0]  # force RLE run
        streams.append(_encode_hybrid(vals, bit_width))
        expected.append(vals)
    cat = b''.join(streams)
    starts = np.zeros(len(streams), dtype=np.int64)
    ends = np.zeros(len(streams), dtype=np.int64)
    pos = 0
    for i, s in enumerate(streams):
        starts[i] = pos
        pos += len(s)
        ends[i] = pos
    nvals = np.array([len(e) for e in expected], dtype=np.int32)
    out_off = np.zeros(len(streams), dtype=np.int64)
    out_off[1:] = np.cumsum(nvals)[:-1]

    dev = 'cuda'
    data = torch.frombuffer(bytearray(cat + b'\0' * 16),
                            dtype=torch.uint8).to(dev)
    out = torch.zeros(int(nvals.sum()), dtype=torch.int32, device=dev)
    status = torch.zeros(len(streams), dtype=torch.int32, device=dev)
    ext.rle_hybrid_decode_batch(
        data, torch.from_numpy(starts).to(dev), torch.from_numpy(ends).to(dev),
        torch.full((len(streams),), bit_width, dtype=torch.int32, device=dev),
        torch.from_numpy(nvals).to(dev), torch.from_numpy(out_off).to(dev),
        out, status)
    torch.cuda.synchronize()
    assert status.cpu().tolist() == [0] * len(streams)
    got = out.cpu().numpy()
    exp = np.concatenate(expected)
    np.testing.assert_array_equal(got, exp)


# ---------------------------------------------------------------------------
# varlen gather
# ---------------------------------------------------------------------------

def test_varlen_gather_misaligned(ext):
    rng = np.random.RandomState(1)
    src = rng.randint(0, 255, 300000).astype(np.uint8)
    items = []
    pos = 1  # deliberately odd start
    while pos + 2000 < len(src):
        ln = int(rng.randint(1, 1999))
        items.append((pos, ln))
        pos += ln + int(rng.randint(0, 3))
    dst_off = np.zeros(len(items), dtype=np.int64)
    lens = np.array([ln for _, ln in items], dtype=np.int64)
    dst_off[1:] = np.cumsum(lens)[:-1]
    dev = 'cuda'
    src_t = torch.from_numpy(src).to(dev)
    dst = torch.zeros(int(lens.sum()) + 16, dtype=torch.uint8, device=dev)
    ext.varlen_gather(src_t,
                      torch.tensor([p for p, _ in items],
                                   dtype=torch.int64, device=dev),
                      torch.from_numpy(lens).to(dev), dst,
                      torch.from_numpy(dst_off).to(dev))
    torch.cuda.synchronize()
    expected = np.concatenate([src[p:p + ln] for p, ln in items])
    np.testing.assert_array_equal(dst[:len(expected)].cpu().numpy(), expected)


# ---------------------------------------------------------------------------
# full rowgroup decode vs pyarrow (the CPU oracle)
# ---------------------------------------------------------------------------

@pytest.mark.parametrize('compression', ['snappy', 'none', 'lz4', 'zstd'])
def test_scalar_rowgroup_decode_vs_pyarrow(ext, tmp_path, compression):
    import pyarrow.parquet as pq
    from petastorm_amd.test_util.dataset_gen import create_scalar_dataset
    from petastorm_amd.gpu.decoder import GpuRowGroupDecoder
    from petastorm_amd.etl import dataset_metadata as dsm
    from petastorm_amd.fs_utils import get_filesystem_and_path_or_paths

    url = 'file://' + str(tmp_path / ('ds_' + compression))
    create_scalar_dataset(url, num_rows=5000, rowgroup_size=1024,
                          compression=compression)
    fs, path = get_filesystem_and_path_or_paths(url)
    pieces = dsm.load_row_groups(fs, path)
    schema, _ = dsm.infer_or_load_unischema(fs, path)
    dec = GpuRowGroupDecoder('cuda')
    cols = ['id', 'f0', 'f3', 'i2']
    for piece in pieces[:3]:
        pf = pq.ParquetFile(piece.path)
        host, meta = dec.read_rowgroup_bytes(piece.path, pf.metadata,
                                             pf.schema, piece.row_group, cols)
        out, _ = dec.decode(host, meta, schema)
        oracle = pf.read_row_group(piece.row_group, columns=cols)
        for c in cols:
            got = out[c].cpu().numpy()
            exp = oracle.column(c).to_numpy()
            np.testing.assert_array_equal(got, exp, err_msg=c)


def test_nullable_rowgroup_decode(ext, tmp_path):
    import pyarrow as pa
    import pyarrow.parquet as pq
    from petastorm_amd.gpu.decoder import GpuRowGroupDecoder
    from petastorm_amd.unischema import Unischema

    rng = np.random.RandomState(0)
    vals = rng.rand(4000)
    mask = rng.rand(4000) < 0.3
    col = pa.array([None if m else float(v) for m, v in zip(mask, vals)],
                   type=pa.float64())
    ids = pa.array(np.arange(4000, dtype=np.int64))
    path = str(tmp_path / 'nullable.parquet')
    pq.write_table(pa.table({'id': ids, 'x': col}), path,
                   compression='snappy', use_dictionary=False,
                   row_group_size=1500)
    pf = pq.ParquetFile(path)
    schema = Unischema.from_arrow_schema(pf.schema_arrow)
    dec = GpuRowGroupDecoder('cuda')
    for rg in range(pf.metadata.num_row_groups):
        host, meta = dec.read_rowgroup_bytes(path, pf.metadata, pf.schema,
                                             rg, ['id', 'x'])
        out, _ = dec.decode(host, meta, schema)
        oracle = pf.read_row_group(rg, columns=['id', 'x'])
        exp = oracle.column('x').to_numpy(zero_copy_only=False)
        got = out['x'].cpu().numpy()
        np.testing.assert_array_equal(np.isnan(got), np.isnan(exp))
        np.testing.assert_allclose(got[~np.isnan(got)], exp[~np.isnan(exp)])


def test_dictionary_encoded_decode(ext, tmp_path):
    import pyarrow as pa
    import pyarrow.parquet as pq
    from petastorm_amd.gpu.decoder import GpuRowGroupDecoder
    from petastorm_amd.unischema import Unischema

    rng = np.random.RandomState(0)
    vals = rng.randint(0, 50, 10000).astype(np.int64)  # few distinct values
    path = str(tmp_path / 'dict.parquet')
    pq.write_table(pa.table({'v': vals}), path, compression='snappy',
                   use_dictionary=True, row_group_size=4000)
    pf = pq.ParquetFile(path)
    schema = Unischema.from_arrow_schema(pf.schema_arrow)
    dec = GpuRowGroupDecoder('cuda')
    for rg in range(pf.metadata.num_row_groups):
        host, meta = dec.read_rowgroup_bytes(path, pf.metadata, pf.schema,
                                             rg, ['v'])
        out, _ = dec.decode(host, meta, schema)
        exp = pf.read_row_group(rg, columns=['v']).column('v').to_numpy()
        np.testing.assert_array_equal(out['v'].cpu().numpy(), exp)


# ---------------------------------------------------------------------------
# ndarray (npy) column
# ---------------------------------------------------------------------------

def test_ndarray_column_decode(ext, tmp_path):
    from petastorm_amd.test_util.dataset_gen import (SequenceSchema,
                                                     create_sequence_dataset)
    from petastorm_amd.gpu.decoder import GpuRowGroupDecoder, ByteArrayColumn
    from petastorm_amd.etl import dataset_metadata as dsm
    from petastorm_amd.fs_utils import get_filesystem_and_path_or_paths
    import pyarrow.parquet as pq

    url = 'file://' + str(tmp_path / 'seq')
    create_sequence_dataset(url, num_rows=100, rowgroup_size_mb=0.5)
    fs, path = get_filesystem_and_path_or_paths(url)
    pieces = dsm.load_row_groups(fs, path)
    schema = dsm.get_schema(fs, path)
    dec = GpuRowGroupDecoder('cuda')
    piece = pieces[0]
    pf = pq.ParquetFile(piece.path)
    host, meta = dec.read_rowgroup_bytes(piece.path, pf.metadata, pf.schema,
                                         piece.row_group,
                                         ['timestamp', 'tokens'])
    out, _ = dec.decode(host, meta, schema)
    col = out['tokens']
    assert isinstance(col, ByteArrayColumn)
    decoded = dec.decode_ndarray_column(col, schema.fields['tokens'])
    assert decoded.shape[1:] == (1024,)
    # oracle: CPU codec decode
    oracle = pf.read_row_group(piece.row_group, columns=['tokens'])
    import io as _io
    exp = np.stack([np.load(_io.BytesIO(v.as_py()))
                    for v in oracle.column('tokens')])
    np.testing.assert_array_equal(decoded.cpu().numpy(), exp)


# ---------------------------------------------------------------------------
# jpeg decode vs PIL
# ---------------------------------------------------------------------------

def _jpeg_batch_to_gpu(ext, blobs):
    dev = 'cuda'
    buf = b''.join(blobs)
    off, lens, pos = [], [], 0
    for d in blobs:
        off.append(pos)
        lens.append(len(d))
        pos += len(d)
    host = torch.frombuffer(bytearray(buf + b'\0' * 16), dtype=torch.uint8)
    meta = ext.jpeg_parse_batch(host, torch.tensor(off, dtype=torch.int64),
                                torch.tensor(lens, dtype=torch.int64))
    dbuf = host.to(dev)
    meta_dev = {k: (v.to(dev) if isinstance(v, torch.Tensor) else v)
                for k, v in meta.items()}
    n = len(blobs)
    widths = meta['width'].numpy()
    heights = meta['height'].numpy()
    ncomp = meta['ncomp'].numpy()
    coef = torch.zeros(int(meta['block_total']) * 64, dtype=torch.float32,
                       device=dev)
    samples = torch.empty(int(meta['samp_total']), dtype=torch.uint8,
                          device=dev)
    out_bytes = widths.astype(np.int64) * heights * np.where(ncomp == 3, 3, 1)
    out_off = np.zeros(n, dtype=np.int64)
    out_off[1:] = np.cumsum(out_bytes)[:-1]
    out = torch.empty(int(out_bytes.sum()), dtype=torch.uint8, device=dev)
    status = torch.zeros(max(1, int(meta['seg_img'].numel())),
                         dtype=torch.int32, device=dev)
    ext.jpeg_decode_batch(dbuf, meta_dev, coef, samples, out,
                          torch.from_numpy(out_off).to(dev), status)
    torch.cuda.synchronize()
    assert int(status.abs().sum()) == 0, status.cpu()
    outs = []
    for i in range(n):
        c = 3 if ncomp[i] == 3 else 1
        img = out[out_off[i]:out_off[i] + out_bytes[i]].cpu().numpy()
        img = img.reshape(heights[i], widths[i], c)
        outs.append(img.squeeze(-1) if c == 1 else img)
    return outs


def _pil_decode(blobs):
    from PIL import Image
    return [np.asarray(Image.open(io.BytesIO(b))) for b in blobs]


def _make_jpegs(n, size=(48, 64), quality=90, gray=False, subsampling=None,
                smooth=False, seed=0):
    from PIL import Image
    rng = np.random.RandomState(seed)
    blobs = []
    h, w = size[1], size[0]
    for i in range(n):
        if smooth:
            yy, xx = np.mgrid[0:h, 0:w].astype(np.float32)
            base = np.sin(xx / 7 + i) * 60 + np.cos(yy / 9) * 50 + 128
            if gray:
                arr = np.clip(base, 0, 255).astype(np.uint8)
            else:
                arr = np.clip(np.stack([base, base * 0.8, 255 - base],
                                       axis=-1), 0, 255).astype(np.uint8)
        elif gray:
            arr = rng.randint(0, 255, (h, w)).astype(np.uint8)
        else:
            arr = rng.randint(0, 255, (h, w, 3)).astype(np.uint8)
        img = Image.fromarray(arr)
        b = io.BytesIO()
        kw = dict(format='JPEG', quality=quality, restart_marker_rows=1)
        if subsampling is not None:
            kw['subsampling'] = subsampling
        img.save(b, **kw)
        blobs.append(b.getvalue())
    return blobs


@pytest.mark.parametrize('subsampling,gray', [
    (0, False),   # 4:4:4
    (2, False),   # 4:2:0
    (1, False),   # 4:2:2
    (None, True),  # grayscale
])
def test_jpeg_decode_matches_pil(ext, subsampling, gray):
    blobs = _make_jpegs(6, size=(48, 64), gray=gray, subsampling=subsampling,
                        smooth=True)
    got = _jpeg_batch_to_gpu(ext, blobs)
    exp = _pil_decode(blobs)
    for g, e in zip(got, exp):
        assert g.shape == e.shape
        diff = np.abs(g.astype(int) - e.astype(int))
        assert diff.mean() < 1.5, diff.mean()
        assert diff.max() <= 8, diff.max()


def test_jpeg_decode_random_noise_image(ext):
    # noise stresses the Huffman decoder (max bitstream entropy)
    blobs = _make_jpegs(4, size=(96, 80), quality=95, subsampling=0)
    got = _jpeg_batch_to_gpu(ext, blobs)
    exp = _pil_decode(blobs)
    for g, e in zip(got, exp):
        diff = np.abs(g.astype(int) - e.astype(int))
        assert diff.mean() < 2.0
        assert diff.max() <= 12


def test_jpeg_decode_odd_sizes(ext):
    # sizes not multiples of the MCU: 50x35 (4:2:0 -> partial MCUs)
    blobs = _make_jpegs(3, size=(35, 50), subsampling=2, smooth=True)
    got = _jpeg_batch_to_gpu(ext, blobs)
    exp = _pil_decode(blobs)
    for g, e in zip(got, exp):
        assert g.shape == e.shape
        assert np.abs(g.astype(int) - e.astype(int)).mean() < 2.0


# ---------------------------------------------------------------------------
# nhwc -> nchw normalize
# ---------------------------------------------------------------------------

def test_nhwc_to_nchw_normalize(ext):
    rng = np.random.RandomState(0)
    x = torch.from_numpy(
        rng.randint(0, 255, (4, 37, 53, 3)).astype(np.uint8)).cuda()
    mean = torch.tensor([0.485, 0.456, 0.406], device='cuda')
    std = torch.tensor([0.229, 0.224, 0.225], device='cuda')
    out = torch.empty(4, 3, 37, 53, dtype=torch.float32, device='cuda')
    ext.nhwc_to_nchw_normalize(x, out, mean, 1.0 / std, 1.0 / 255.0)
    torch.cuda.synchronize()
    ref = (x.permute(0, 3, 1, 2).float() / 255.0 -
           mean.view(1, 3, 1, 1)) / std.view(1, 3, 1, 1)
    torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-5)


def test_nhwc_to_nchw_normalize_bf16(ext):
    rng = np.random.RandomState(0)
    x = torch.from_numpy(
        rng.randint(0, 255, (2, 64, 64, 3)).astype(np.uint8)).cuda()
    mean = torch.zeros(3, device='cuda')
    inv_std = torch.ones(3, device='cuda')
    out = torch.empty(2, 3, 64, 64, dtype=torch.bfloat16, device='cuda')
    ext.nhwc_to_nchw_normalize(x, out, mean, inv_std, 1.0 / 255.0)
    torch.cuda.synchronize()
    ref = (x.permute(0, 3, 1, 2).float() / 255.0).bfloat16()
    torch.testing.assert_close(out.float(), ref.float(), rtol=0.02,
                               atol=0.01)


# ---------------------------------------------------------------------------
# GpuBatchReader end-to-end
# ---------------------------------------------------------------------------

def test_gpu_batch_reader_imagenet(ext, tmp_path):
    from petastorm_amd import make_batch_reader
    from petastorm_amd.test_util.dataset_gen import create_imagenet_dataset
    url = 'file://' + str(tmp_path / 'imnet')
    create_imagenet_dataset(url, num_rows=64, rowgroup_size_mb=8)
    with make_batch_reader(url, device='cuda',
                           shuffle_row_groups=False) as r:
        batches = list(r)
    total = sum(b.image.shape[0] for b in batches)
    assert total == 64
    b0 = batches[0]
    assert b0.image.is_cuda and b0.image.dtype == torch.uint8
    assert b0.image.shape[1:] == (224, 224, 3)
    assert b0.label.is_cuda
    # decode correctness vs CPU reader
    from petastorm_amd import make_reader
    with make_reader(url, reader_pool_type='dummy',
                     shuffle_row_groups=False) as cr:
        cpu_rows = {int(row.label): row.image for row in cr}
    labels = torch.cat([b.label for b in batches]).cpu().numpy()
    images = torch.cat([b.image for b in batches]).cpu().numpy()
    for i in range(0, 64, 16):
        diff = np.abs(images[i].astype(int) -
                      cpu_rows[int(labels[i])].astype(int))
        assert diff.mean() < 2.0


def test_gpu_batch_reader_scalar_sharded(ext, tmp_path):
    from petastorm_amd import make_batch_reader
    from petastorm_amd.test_util.dataset_gen import create_scalar_dataset
    url = 'file://' + str(tmp_path / 'scal')
    create_scalar_dataset(url, num_rows=2000, rowgroup_size=250)
    all_ids = []
    for shard in range(2):
        with make_batch_reader(url, device='cuda', shuffle_row_groups=False,
                               cur_shard=shard, shard_count=2,
                               schema_fields=['id', 'f0']) as r:
            for b in r:
                all_ids.extend(b.id.cpu().tolist())
    assert sorted(all_ids) == list(range(2000))


def test_gpu_batch_reader_hbm_cache_and_epochs(ext, tmp_path):
    from petastorm_amd import make_batch_reader
    from petastorm_amd.test_util.dataset_gen import create_sequence_dataset
    url = 'file://' + str(tmp_path / 'seqc')
    create_sequence_dataset(url, num_rows=64, rowgroup_size_mb=0.5)
    with make_batch_reader(url, device='cuda', shuffle_row_groups=False,
                           num_epochs=3,
                           gpu_options=dict(cache_type='hbm',
                                            cache_size_limit=1 << 30)) as r:
        batches = list(r)
        diag = r.diagnostics
    total = sum(b.tokens.shape[0] for b in batches)
    assert total == 3 * 64
    assert diag['hbm_cache_hits'] > 0


def test_gpu_predicate(ext, tmp_path):
    from petastorm_amd import make_batch_reader
    from petastorm_amd.predicates import in_lambda
    from petastorm_amd.test_util.dataset_gen import create_scalar_dataset
    url = 'file://' + str(tmp_path / 'pred')
    create_scalar_dataset(url, num_rows=1000, rowgroup_size=200)
    pred = in_lambda(['id'], lambda v: v['id'] % 4 == 0)
    with make_batch_reader(url, device='cuda', shuffle_row_groups=False,
                           predicate=pred,
                           schema_fields=['id', 'i1']) as r:
        ids = torch.cat([b.id for b in r]).cpu().numpy()
    assert (ids % 4 == 0).all() and len(ids) == 250


# ---------------------------------------------------------------------------
# inflate / png / compressed-ndarray
# ---------------------------------------------------------------------------

def test_inflate_batch_zlib(ext):
    import zlib
    rng = np.random.RandomState(0)
    payloads = [
        rng.randint(0, 255, 50000).astype(np.uint8).tobytes(),  # stored-ish
        (b'hello world ' * 5000),                               # dynamic huff
        bytes(100000),                                          # zeros
        b'a',                                                   # tiny
    ]
    comp = [zlib.compress(p, 6) for p in payloads]
    cat = b''.join(comp)
    seg_off, seg_len, pos = [], [], 0
    for c in comp:
        seg_off.append(pos)
        seg_len.append(len(c))
        pos += len(c)
    dev = 'cuda'
    src = torch.frombuffer(bytearray(cat + b'\0' * 16),
                           dtype=torch.uint8).to(dev)
    caps = np.array([len(p) for p in payloads], dtype=np.int64)
    dst_off = np.zeros(len(payloads), dtype=np.int64)
    dst_off[1:] = np.cumsum(caps)[:-1]
    dst = torch.zeros(int(caps.sum()) + 16, dtype=torch.uint8, device=dev)
    produced = torch.zeros(len(payloads), dtype=torch.int64, device=dev)
    status = torch.zeros(len(payloads), dtype=torch.int32, device=dev)
    ext.inflate_batch(
        src, torch.tensor(seg_off, dtype=torch.int64, device=dev),
        torch.tensor(seg_len, dtype=torch.int64, device=dev),
        torch.arange(len(payloads), dtype=torch.int32, device=dev),
        torch.ones(len(payloads), dtype=torch.int32, device=dev),
        dst, torch.from_numpy(dst_off).to(dev),
        torch.from_numpy(caps).to(dev), produced, 0, status)
    torch.cuda.synchronize()
    assert status.cpu().tolist() == [0] * len(payloads)
    assert produced.cpu().numpy().tolist() == [len(p) for p in payloads]
    got = dst[:int(caps.sum())].cpu().numpy().tobytes()
    assert got == b''.join(payloads)


@pytest.mark.parametrize('mode', ['rgb', 'gray', 'gray16', 'rgba'])
def test_png_column_decode(ext, tmp_path, mode):
    from PIL import Image
    from petastorm_amd.gpu.decoder import GpuRowGroupDecoder
    rng = np.random.RandomState(0)
    n = 5
    blobs, arrays = [], []
    for i in range(n):
        if mode == 'rgb':
            arr = rng.randint(0, 255, (40, 30, 3)).astype(np.uint8)
        elif mode == 'gray':
            arr = rng.randint(0, 255, (40, 30)).astype(np.uint8)
        elif mode == 'gray16':
            arr = rng.randint(0, 2 ** 16, (40, 30)).astype(np.uint16)
        else:
            arr = rng.randint(0, 255, (40, 30, 4)).astype(np.uint8)
        img = Image.fromarray(arr)
        b = io.BytesIO()
        img.save(b, format='PNG')
        blobs.append(b.getvalue())
        arrays.append(arr)
    # exercise the codec-kernel path directly through a synthetic column
    dev = 'cuda'
    buf = b''.join(blobs)
    off, lens, pos = [], [], 0
    for d in blobs:
        off.append(pos)
        lens.append(len(d))
        pos += len(d)
    host = torch.frombuffer(bytearray(buf + b'\0' * 16), dtype=torch.uint8)
    from petastorm_amd.gpu.decoder import ByteArrayColumn
    from petastorm_amd.unischema import UnischemaField
    from petastorm_amd.codecs import CompressedImageCodec
    col = ByteArrayColumn(host.to(dev),
                          torch.tensor(off, dtype=torch.int64, device=dev),
                          torch.tensor(lens, dtype=torch.int32, device=dev),
                          host, np.array(off, dtype=np.int64), n)
    dtype = np.uint16 if mode == 'gray16' else np.uint8
    shape = arrays[0].shape
    field = UnischemaField('im', dtype, shape, CompressedImageCodec('png'),
                           False)
    dec = GpuRowGroupDecoder(dev)
    out = dec.decode_png_column(col, field)
    dec.flush_status()
    torch.cuda.synchronize()
    assert out is not None
    got = out.cpu().numpy()
    exp = np.stack(arrays)
    np.testing.assert_array_equal(got.astype(np.int64), exp.astype(np.int64))


def test_compressed_ndarray_column_via_reader(ext, tmp_path):
    from petastorm_amd import make_reader, make_batch_reader
    from petastorm_amd.codecs import CompressedNdarrayCodec, ScalarCodec
    from petastorm_amd.etl.dataset_metadata import materialize_dataset
    from petastorm_amd.unischema import Unischema, UnischemaField
    schema = Unischema('Z', [
        UnischemaField('id', np.int64, (), ScalarCodec(), False),
        UnischemaField('mat', np.float32, (32, 16), CompressedNdarrayCodec(),
                       False),
    ])
    url = 'file://' + str(tmp_path / 'zds')
    rng = np.random.RandomState(0)
    rows = [{'id': np.int64(i), 'mat': rng.rand(32, 16).astype(np.float32)}
            for i in range(50)]
    with materialize_dataset(url, schema, 1) as w:
        w.write_rows(rows)
    with make_batch_reader(url, device='cuda', shuffle_row_groups=False) as r:
        batches = list(r)
        assert not r.diagnostics['cpu_assist_columns']
    ids = torch.cat([b.id for b in batches]).cpu().numpy()
    mats = torch.cat([b.mat for b in batches]).cpu().numpy()
    by_id = {int(i): m for i, m in zip(ids, mats)}
    for src in rows:
        np.testing.assert_allclose(by_id[int(src['id'])], src['mat'],
                                   rtol=1e-6)


def test_gpu_batch_reader_helloworld_png(ext, tmp_path):
    """HelloWorld schema (png images + 4-D ndarray) through the GPU path."""
    from petastorm_amd import make_batch_reader, make_reader
    from petastorm_amd.test_util.dataset_gen import create_hello_world_dataset
    url = 'file://' + str(tmp_path / 'hw')
    create_hello_world_dataset(url, num_rows=24, rowgroup_size_mb=8)
    with make_batch_reader(url, device='cuda', shuffle_row_groups=False) as r:
        batches = list(r)
        assist = r.diagnostics['cpu_assist_columns']
    # array_4d has variable shape -> allowed to take the CPU assist;
    # image1 (png) and id must be native
    assert 'image1' not in assist and 'id' not in assist
    total = sum(b.id.shape[0] for b in batches)
    assert total == 24
    with make_reader(url, reader_pool_type='dummy',
                     shuffle_row_groups=False) as cr:
        cpu = {int(row.id): row.image1 for row in cr}
    for b in batches:
        ids = b.id.cpu().numpy()
        imgs = b.image1.cpu().numpy()
        for i in range(0, len(ids), 8):
            np.testing.assert_array_equal(imgs[i], cpu[int(ids[i])])


def test_gpu_reader_state_dict_resume(ext, tmp_path):
    """Checkpoint/resume: a reloaded reader continues at the saved cursor."""
    from petastorm_amd import make_batch_reader
    from petastorm_amd.test_util.dataset_gen import create_scalar_dataset
    url = 'file://' + str(tmp_path / 'ckpt')
    create_scalar_dataset(url, num_rows=1000, rowgroup_size=100)
    kwargs = dict(device='cuda', shuffle_row_groups=True, seed=99,
                  num_epochs=2, schema_fields=['id'])
    with make_batch_reader(url, **kwargs) as r1:
        it = iter(r1)
        seen = [next(it) for _ in range(4)]
        state = r1.state_dict()
        rest_a = [b.id.cpu() for b in it]
    with make_batch_reader(url, **kwargs) as r2:
        r2.load_state_dict(state)
        rest_b = [b.id.cpu() for b in r2]
    assert len(rest_a) == len(rest_b)
    for a, b in zip(rest_a, rest_b):
        assert torch.equal(a, b)


def test_gzip_rowgroup_decode(ext, tmp_path):
    from petastorm_amd.test_util.dataset_gen import create_scalar_dataset
    from petastorm_amd import make_batch_reader
    url = 'file://' + str(tmp_path / 'gz')
    create_scalar_dataset(url, num_rows=3000, rowgroup_size=1000,
                          compression='gzip')
    with make_batch_reader(url, device='cuda', shuffle_row_groups=False,
                           schema_fields=['id', 'f0']) as r:
        ids = torch.cat([b.id for b in r]).cpu().numpy()
        assert not r.diagnostics['cpu_assist_columns']
    np.testing.assert_array_equal(np.sort(ids), np.arange(3000))


@pytest.mark.parametrize('compression', ['snappy', 'none', 'gzip', 'zstd', 'lz4'])
def test_datapage_v2_decode(ext, tmp_path, compression):
    import pyarrow as pa
    import pyarrow.parquet as pq
    from petastorm_amd.gpu.decoder import GpuRowGroupDecoder
    from petastorm_amd.unischema import Unischema
    rng = np.random.RandomState(0)
    vals = rng.rand(5000)
    mask = rng.rand(5000) < 0.2
    col = pa.array([None if m else float(v) for m, v in zip(mask, vals)])
    path = str(tmp_path / ('v2_' + compression + '.parquet'))
    pq.write_table(pa.table({'id': pa.array(np.arange(5000)), 'x': col}),
                   path, compression=compression, use_dictionary=False,
                   row_group_size=2000, data_page_version='2.0')
    pf = pq.ParquetFile(path)
    schema = Unischema.from_arrow_schema(pf.schema_arrow)
    dec = GpuRowGroupDecoder('cuda')
    for rg in range(pf.metadata.num_row_groups):
        host, meta = dec.read_rowgroup_bytes(path, pf.metadata, pf.schema,
                                             rg, ['id', 'x'])
        out, _ = dec.decode(host, meta, schema)
        dec.flush_status()
        assert not dec.cpu_assist_columns, dec.cpu_assist_columns
        oracle = pf.read_row_group(rg, columns=['id', 'x'])
        got_x = out['x'].cpu().numpy()
        exp_x = oracle.column('x').to_numpy(zero_copy_only=False)
        np.testing.assert_array_equal(np.isnan(got_x), np.isnan(exp_x))
        np.testing.assert_allclose(got_x[~np.isnan(got_x)],
                                   exp_x[~np.isnan(exp_x)])
        np.testing.assert_array_equal(out['id'].cpu().numpy(),
                                      oracle.column('id').to_numpy())


def test_dictionary_byte_array_ndarray_decode(ext, tmp_path):
    """Dictionary-encoded binary column (repeated blobs) decoded on GPU."""
    import io as _io
    import pyarrow as pa
    import pyarrow.parquet as pq
    from petastorm_amd.gpu.decoder import ByteArrayColumn, GpuRowGroupDecoder
    from petastorm_amd.unischema import Unischema, UnischemaField
    from petastorm_amd.codecs import NdarrayCodec
    rng = np.random.RandomState(0)
    distinct = []
    for _ in range(20):
        buf = _io.BytesIO()
        np.save(buf, rng.rand(16, 8).astype(np.float32))
        distinct.append(buf.getvalue())
    blobs = [distinct[i % 20] for i in range(400)]
    path = str(tmp_path / 'dictba.parquet')
    pq.write_table(pa.table({'mat': pa.array(blobs, type=pa.binary())}),
                   path, compression='snappy', use_dictionary=True,
                   row_group_size=200)
    pf = pq.ParquetFile(path)
    # confirm it actually dictionary-encoded
    encs = pf.metadata.row_group(0).column(0).encodings
    assert any('DICTIONARY' in e for e in encs), encs
    schema = Unischema('D', [UnischemaField('mat', np.float32, (16, 8),
                                            NdarrayCodec(), False)])
    dec = GpuRowGroupDecoder('cuda')
    for rg in range(pf.metadata.num_row_groups):
        host, meta = dec.read_rowgroup_bytes(path, pf.metadata, pf.schema,
                                             rg, ['mat'])
        out, _ = dec.decode(host, meta, schema)
        col = out['mat']
        assert isinstance(col, ByteArrayColumn)
        decoded = dec.decode_ndarray_column(col, schema.fields['mat'])
        dec.flush_status()
        torch.cuda.synchronize()
        oracle = pf.read_row_group(rg, columns=['mat']).column('mat')
        exp = np.stack([np.load(_io.BytesIO(v.as_py())) for v in oracle])
        np.testing.assert_array_equal(decoded.cpu().numpy(), exp)


def test_lz4_kernel_direct(ext):
    """Handwritten LZ4 blocks: literals, a long run-replicating match, and
    multi-byte length extensions; plus a short-offset (off < lanes) match."""
    dev = torch.device('cuda')
    rng = np.random.RandomState(7)

    # block A: 8 literals + matchlen-84 run copy (offset 8) + 3 literals
    expected_a = b'abcdefgh' * 11 + b'XYZ'   # 8 + 80 + 3
    blk_a = bytes([0x8F]) + b'abcdefgh' + b'\x08\x00' + bytes([80 - 4 - 15])
    blk_a += bytes([0x30]) + b'XYZ'

    # block B: 300 literals only (length extension 15+255+30)
    lits = rng.randint(0, 256, 300, dtype=np.uint8).tobytes()
    blk_b = bytes([0xF0, 255, 30]) + lits
    expected_b = lits

    # block C: offset 3 < wave width -> i % off replication path
    expected_c = b'abc' * 40 + b'Q'          # 3 + 117 + 1
    blk_c = bytes([0x3F]) + b'abc' + b'\x03\x00' + bytes([117 - 4 - 15])
    blk_c += bytes([0x10]) + b'Q'

    comp = torch.from_numpy(np.frombuffer(
        blk_a + blk_b + blk_c, dtype=np.uint8).copy()).to(dev)
    starts = torch.tensor([0, len(blk_a), len(blk_a) + len(blk_b)],
                          dtype=torch.int64, device=dev)
    ends = torch.tensor([len(blk_a), len(blk_a) + len(blk_b),
                         len(blk_a) + len(blk_b) + len(blk_c)],
                        dtype=torch.int64, device=dev)
    lens = [len(expected_a), len(expected_b), len(expected_c)]
    out_off = torch.tensor([0, lens[0], lens[0] + lens[1]],
                           dtype=torch.int64, device=dev)
    out = torch.empty(sum(lens) + 16, dtype=torch.uint8, device=dev)
    status = torch.zeros(3, dtype=torch.int32, device=dev)
    ext.lz4_decompress_batch(comp, starts, ends, out,
                             out_off, torch.tensor(lens, dtype=torch.int64,
                                                   device=dev), status)
    torch.cuda.synchronize()
    assert status.cpu().tolist() == [0, 0, 0]
    got = bytes(out.cpu().numpy().tobytes())
    assert got[:lens[0]] == expected_a
    assert got[lens[0]:lens[0] + lens[1]] == expected_b
    assert got[lens[0] + lens[1]:sum(lens)] == expected_c


def test_lz4_reader_end_to_end(ext, tmp_path):
    from petastorm_amd.test_util.dataset_gen import create_scalar_dataset
    from petastorm_amd import make_batch_reader
    url = 'file://' + str(tmp_path / 'lz4')
    create_scalar_dataset(url, num_rows=3000, rowgroup_size=1000,
                          compression='lz4')
    with make_batch_reader(url, device='cuda', shuffle_row_groups=False,
                           schema_fields=['id', 'f0']) as r:
        ids = torch.cat([b.id for b in r]).cpu().numpy()
        assert not r.diagnostics['cpu_assist_columns']
    np.testing.assert_array_equal(np.sort(ids), np.arange(3000))


def test_gpu_ngram_reader_matches_cpu(ext, tmp_path):
    """make_batch_reader(schema_fields=NGram, device='cuda') assembles the
    same windows as the CPU make_reader NGram path (reference
    ngram.py:225-270), batched as {timestep: namedtuple-of-tensors}."""
    from petastorm_amd import make_reader, make_batch_reader
    from petastorm_amd.ngram import NGram
    from petastorm_amd.test_util.dataset_gen import create_sequence_dataset

    url = 'file://' + str(tmp_path / 'seq')
    create_sequence_dataset(url, num_rows=600, rows_per_rowgroup=200)

    def make_ngram():
        return NGram(fields={0: ['timestamp', 'source', 'tokens'],
                             1: ['timestamp', 'source']},
                     delta_threshold=1, timestamp_field='timestamp')

    cpu_windows = {}
    with make_reader(url, schema_fields=make_ngram(),
                     reader_pool_type='dummy', shuffle_row_groups=False,
                     num_epochs=1) as r:
        for w in r:
            cpu_windows[int(w[0].timestamp)] = (
                int(w[1].timestamp), int(w[0].source), int(w[1].source),
                np.asarray(w[0].tokens))

    gpu_windows = {}
    with make_batch_reader(url, schema_fields=make_ngram(), device='cuda',
                           shuffle_row_groups=False, num_epochs=1) as r:
        for batch in r:
            t0, t1 = batch[0], batch[1]
            ts0 = t0.timestamp.cpu().numpy()
            ts1 = t1.timestamp.cpu().numpy()
            s0 = t0.source.cpu().numpy()
            s1 = t1.source.cpu().numpy()
            toks = t0.tokens.cpu().numpy()
            for i in range(len(ts0)):
                gpu_windows[int(ts0[i])] = (int(ts1[i]), int(s0[i]),
                                            int(s1[i]), toks[i])

    assert set(gpu_windows) == set(cpu_windows)
    for k, (t1c, s0c, s1c, tokc) in cpu_windows.items():
        t1g, s0g, s1g, tokg = gpu_windows[k]
        assert (t1c, s0c, s1c) == (t1g, s0g, s1g)
        np.testing.assert_array_equal(tokc, tokg)


def test_zstd_reader_end_to_end(ext, tmp_path):
    from petastorm_amd.test_util.dataset_gen import create_scalar_dataset
    from petastorm_amd import make_batch_reader
    url = 'file://' + str(tmp_path / 'zstd')
    create_scalar_dataset(url, num_rows=3000, rowgroup_size=1000,
                          compression='zstd')
    with make_batch_reader(url, device='cuda', shuffle_row_groups=False,
                           schema_fields=['id', 'f0']) as r:
        ids = torch.cat([b.id for b in r]).cpu().numpy()
        assert not r.diagnostics['cpu_assist_columns']
    np.testing.assert_array_equal(np.sort(ids), np.arange(3000))


def test_zstd_device_kernel_matches_host(ext, tmp_path):
    """Thread-per-frame GPU zstd kernel vs the host decoder on
    libzstd-compressed data."""
    import ctypes
    z = ctypes.CDLL('libzstd.so.1')
    z.ZSTD_compress.restype = ctypes.c_size_t
    z.ZSTD_compressBound.restype = ctypes.c_size_t
    rng = np.random.RandomState(11)
    datas = [bytes(rng.randint(0, 256, 4000, dtype=np.uint8)),
             bytes(rng.randint(0, 9, 150000, dtype=np.uint8)),
             (b'seq-' * 50000),
             rng.rand(30000).tobytes()]
    comps = []
    for d, lvl in zip(datas, (1, 3, 9, 19)):
        buf = ctypes.create_string_buffer(z.ZSTD_compressBound(len(d)))
        n = z.ZSTD_compress(buf, len(buf.raw), d, len(d), lvl)
        comps.append(buf.raw[:n])
    dev = torch.device('cuda')
    blob = b''.join(comps)
    src = torch.from_numpy(np.frombuffer(blob, dtype=np.uint8).copy()).to(dev)
    soff = np.cumsum([0] + [len(c) for c in comps])[:-1]
    doff = np.cumsum([0] + [len(d) for d in datas])[:-1]
    dlen = [len(d) for d in datas]
    dst = torch.empty(int(sum(dlen)), dtype=torch.uint8, device=dev)
    work = torch.empty(len(datas) * int(ext.zstd_work_bytes()),
                       dtype=torch.uint8, device=dev)
    status = torch.zeros(len(datas), dtype=torch.int32, device=dev)
    ext.zstd_decompress_batch(
        src, torch.tensor(soff, dtype=torch.int64, device=dev),
        torch.tensor([len(c) for c in comps], dtype=torch.int64, device=dev),
        dst, torch.tensor(doff, dtype=torch.int64, device=dev),
        torch.tensor(dlen, dtype=torch.int64, device=dev), work, status)
    torch.cuda.synchronize()
    assert status.cpu().tolist() == [0] * len(datas)
    out = dst.cpu().numpy().tobytes()
    for d, o, n in zip(datas, doff, dlen):
        assert out[o:o + n] == d


def test_corrupt_lz4_page_raises(ext, tmp_path):
    """Corrupt LZ4 page -> kernel status propagates as a loud RuntimeError
    at the reader's status check."""
    import pyarrow.parquet as pq
    from petastorm_amd.test_util.dataset_gen import create_scalar_dataset
    from petastorm_amd import make_batch_reader
    url = 'file://' + str(tmp_path / 'lc')
    create_scalar_dataset(url, num_rows=2000, rowgroup_size=1000,
                          compression='lz4')
    path = [str(p) for p in (tmp_path / 'lc').iterdir()
            if p.suffix == '.parquet'][0]
    pf = pq.ParquetFile(path)
    col = pf.metadata.row_group(0).column(2)
    raw = bytearray(open(path, 'rb').read())
    off = col.data_page_offset + 40
    for i in range(16):
        raw[off + i] ^= 0x5A
    open(path, 'wb').write(bytes(raw))
    with pytest.raises(RuntimeError):
        with make_batch_reader(url, device='cuda',
                               shuffle_row_groups=False) as r:
            for _ in r:
                pass


def test_gpu_reader_hive_partitions(ext, tmp_path):
    """Hive-partitioned dataset on the GPU path: partition columns are
    excluded from the physical read and materialized as device constants
    (string keys stay host-side object arrays)."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    from petastorm_amd import make_batch_reader
    root = tmp_path / 'pds'
    idx = 0
    for color, code in (('red', 0), ('green', 1), ('blue', 2)):
        d = root / 'color={}'.format(color) / 'code={}'.format(code)
        d.mkdir(parents=True)
        pq.write_table(
            pa.table({'id': np.arange(idx, idx + 50, dtype=np.int64)}),
            str(d / 'p.parquet'), use_dictionary=False, row_group_size=25)
        idx += 50
    url = 'file://' + str(root)
    seen = {}
    with make_batch_reader(url, device='cuda',
                           shuffle_row_groups=False) as r:
        assert 'color' in r.schema.fields and 'code' in r.schema.fields
        for b in r:
            assert b.id.is_cuda
            assert b.code.is_cuda           # int partition -> device tensor
            codes = set(b.code.cpu().tolist())
            assert len(codes) == 1
            colors = set(np.asarray(b.color).tolist())
            assert len(colors) == 1
            seen[colors.pop()] = codes.pop()
    assert seen == {'red': 0, 'green': 1, 'blue': 2}


# ---------------------------------------------------------------------------
# device-side strings / nullable BYTE_ARRAY (VERDICT r1 missing item 5)
# ---------------------------------------------------------------------------

@pytest.mark.parametrize('compression', ['snappy', 'none'])
def test_string_column_device_path(ext, tmp_path, compression):
    """String columns decode via GPU page decode + boundary
    materialization: values exact, and cpu_assist_columns == [] for the
    reference-style scalar store (VERDICT Done-condition)."""
    from petastorm_amd import make_batch_reader
    from petastorm_amd.test_util.dataset_gen import create_scalar_dataset
    url = 'file://' + str(tmp_path / ('s_' + compression))
    cols = create_scalar_dataset(url, num_rows=1000, rowgroup_size=250,
                                 compression=compression)
    with make_batch_reader(url, device='cuda', shuffle_row_groups=False,
                           schema_fields=['id', 'f0', 'name']) as r:
        got_ids, got_names = [], []
        for b in r:
            got_ids.append(b.id.cpu().numpy())
            assert isinstance(b.name, np.ndarray)
            got_names.append(b.name)
        assert r.diagnostics['cpu_assist_columns'] == []
    ids = np.concatenate(got_ids)
    names = np.concatenate([np.asarray(g, dtype=object) for g in got_names])
    by_id = {int(i): n for i, n in zip(ids, names)}
    for i in range(1000):
        assert by_id[i] == cols['name'][i]


def test_nullable_string_and_binary_columns_gpu(ext, tmp_path):
    """OPTIONAL BYTE_ARRAY with actual nulls: def levels decode on GPU,
    nulls surface as None in an object array; bytes columns return bytes."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    from petastorm_amd import make_batch_reader
    d = tmp_path / 'nulls'
    d.mkdir()
    n = 500
    names = [None if i % 7 == 0 else 'val-%d' % i for i in range(n)]
    blobs = [None if i % 11 == 0 else bytes([i % 256]) * (i % 17 + 1)
             for i in range(n)]
    table = pa.table({
        'id': pa.array(np.arange(n, dtype=np.int64)),
        'sval': pa.array(names, type=pa.string()),
        'bval': pa.array(blobs, type=pa.binary()),
    })
    pq.write_table(table, str(d / 'p.parquet'), row_group_size=125,
                   compression='snappy', use_dictionary=False)
    with make_batch_reader('file://' + str(d), device='cuda',
                           shuffle_row_groups=False) as r:
        got = {'id': [], 'sval': [], 'bval': []}
        for b in r:
            got['id'].append(b.id.cpu().numpy())
            got['sval'].append(np.asarray(b.sval, dtype=object))
            got['bval'].append(np.asarray(b.bval, dtype=object))
        assert r.diagnostics['cpu_assist_columns'] == []
    ids = np.concatenate(got['id'])
    svals = np.concatenate(got['sval'])
    bvals = np.concatenate(got['bval'])
    for i, rid in enumerate(ids):
        assert svals[i] == names[int(rid)]
        assert bvals[i] == blobs[int(rid)]


def test_foreign_jpeg_no_rst_decodes_on_gpu(ext, tmp_path):
    """Baseline JPEGs WITHOUT restart markers (cv2/PIL defaults — foreign
    datasets): single-segment-per-image GPU decode, values matching the
    CPU (PIL) oracle."""
    import os
    from petastorm_amd import make_batch_reader, make_reader
    from petastorm_amd.test_util.dataset_gen import create_imagenet_dataset
    url = 'file://' + str(tmp_path / 'foreign')
    os.environ['PSA_JPEG_RST_BLOCKS'] = '0'
    try:
        create_imagenet_dataset(url, num_rows=16, rowgroup_size_mb=8)
    finally:
        del os.environ['PSA_JPEG_RST_BLOCKS']
    with make_batch_reader(url, device='cuda',
                           shuffle_row_groups=False) as r:
        batches = list(r)
        assert r.diagnostics['cpu_assist_columns'] == []
    with make_reader(url, reader_pool_type='dummy',
                     shuffle_row_groups=False) as cr:
        cpu = {int(row.label): row.image for row in cr}
    total = 0
    for b in batches:
        labels = b.label.cpu().numpy()
        imgs = b.image.cpu().numpy()
        for lab, img in zip(labels, imgs):
            diff = np.abs(img.astype(np.int16) -
                          cpu[int(lab)].astype(np.int16))
            # same tolerance as the RST-coded jpeg oracle tests above:
            # the GPU IDCT rounds differently from PIL's
            assert diff.mean() < 1.5 and diff.max() <= 8, \
                (diff.mean(), diff.max())
            total += 1
    assert total == 16


def test_dict_encoded_nullable_strings_gpu(ext, tmp_path):
    """Dictionary-encoded string column WITH nulls: def levels + RLE
    indices + dictionary gather on GPU, nulls as None (previously a CPU
    assist)."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    from petastorm_amd import make_batch_reader
    d = tmp_path / 'dictnull'
    d.mkdir()
    n = 600
    cats = ['alpha', 'beta', 'gamma', 'delta']
    vals = [None if i % 9 == 0 else cats[i % 4] for i in range(n)]
    table = pa.table({
        'id': pa.array(np.arange(n, dtype=np.int64)),
        'cat': pa.array(vals, type=pa.string()),
    })
    pq.write_table(table, str(d / 'p.parquet'), row_group_size=150,
                   compression='snappy', use_dictionary=['cat'])
    with make_batch_reader('file://' + str(d), device='cuda',
                           shuffle_row_groups=False) as r:
        got_ids, got_cats = [], []
        for b in r:
            got_ids.append(b.id.cpu().numpy())
            got_cats.append(np.asarray(b.cat, dtype=object))
        assert r.diagnostics['cpu_assist_columns'] == []
    ids = np.concatenate(got_ids)
    catv = np.concatenate(got_cats)
    for i, rid in enumerate(ids):
        assert catv[i] == vals[int(rid)], (rid, catv[i], vals[int(rid)])


def test_decimal_column_gpu_matches_cpu_batch_route(ext, tmp_path):
    """Decimal scalar fields ride the device string path; values match
    the CPU batch route (both surface decimal-as-string on batch
    readers)."""
    from decimal import Decimal
    from petastorm_amd import make_batch_reader
    from petastorm_amd.codecs import ScalarCodec
    from petastorm_amd.etl.dataset_metadata import materialize_dataset
    from petastorm_amd.unischema import Unischema, UnischemaField
    schema = Unischema('D', [
        UnischemaField('id', np.int64, (), ScalarCodec(), False),
        UnischemaField('price', Decimal, (), ScalarCodec(), False)])
    url = 'file://' + str(tmp_path / 'dec')
    with materialize_dataset(url, schema, 1) as w:
        w.write_rows([{'id': np.int64(i),
                       'price': Decimal(i) / Decimal(7)}
                      for i in range(50)])
    with make_batch_reader(url, device='cuda',
                           shuffle_row_groups=False) as r:
        b = next(iter(r))
        assert r.diagnostics['cpu_assist_columns'] == []
        gpu_prices = {int(i): p for i, p in
                      zip(b.id.cpu().numpy(), np.asarray(b.price))}
    with make_batch_reader(url, shuffle_row_groups=False) as cr:
        cb = next(iter(cr))
        cpu_prices = {int(i): p for i, p in
                      zip(np.asarray(cb.id), np.asarray(cb.price))}
    assert set(gpu_prices) == set(cpu_prices)
    for k in gpu_prices:
        assert str(gpu_prices[k]) == str(cpu_prices[k])
        assert Decimal(str(gpu_prices[k])) == Decimal(k) / Decimal(7)


def test_fused_jpeg_normalize_bit_exact(ext, tmp_path):
    """fused_image_normalize through the GPU route must be BIT-EXACT with
    the two-step path (decode to uint8 NHWC, then the standalone
    normalize kernel): the fused epilogue rounds/clamps to u8 first."""
    from petastorm_amd import TransformSpec, make_batch_reader
    from petastorm_amd.transform import fused_image_normalize
    from petastorm_amd.test_util.dataset_gen import create_imagenet_dataset
    from petastorm_amd.unischema import UnischemaField
    url = 'file://' + str(tmp_path / 'fuse')
    create_imagenet_dataset(url, num_rows=24, rowgroup_size_mb=8)
    mean, std = [0.485, 0.456, 0.406], [0.229, 0.224, 0.225]

    ts_fused = fused_image_normalize('image', mean, std)
    with make_batch_reader(url, device='cuda', shuffle_row_groups=False,
                           transform_spec=ts_fused) as r:
        fused = [(b.label.cpu().numpy(), b.image) for b in r]
        assert r.diagnostics['cpu_assist_columns'] == []

    e = ext
    mean_t = torch.tensor(mean, device='cuda')
    inv_t = 1.0 / torch.tensor(std, device='cuda')

    def two_step(cols):
        img = cols['image']
        out = torch.empty(img.shape[0], 3, img.shape[1], img.shape[2],
                          dtype=torch.float32, device=img.device)
        e.nhwc_to_nchw_normalize(img, out, mean_t, inv_t, 1.0 / 255.0)
        return {'image': out, 'label': cols['label']}

    ts_two = TransformSpec(two_step, edit_fields=[
        UnischemaField('image', np.float32, (3, 224, 224), None, False)])
    with make_batch_reader(url, device='cuda', shuffle_row_groups=False,
                           transform_spec=ts_two) as r:
        two = [(b.label.cpu().numpy(), b.image) for b in r]

    assert len(fused) == len(two)
    for (lf, imf), (lt, imt) in zip(fused, two):
        np.testing.assert_array_equal(lf, lt)
        assert imf.shape == imt.shape and imf.dtype == imt.dtype
        torch.testing.assert_close(imf, imt, rtol=0, atol=0)


def test_rich_scalar_store_gpu_route(ext, tmp_path):
    """Full scalar-type surface through the GPU route: ints/floats/strings
    native; date/timestamp values must MATCH the CPU route's mapping
    regardless of which side decodes them."""
    from petastorm_amd import make_batch_reader
    from petastorm_amd.test_util.dataset_gen import create_rich_scalar_dataset
    url = 'file://' + str(tmp_path / 'richg')
    rows = create_rich_scalar_dataset(url, num_rows=60, rowgroup_size=20)

    def read_all(device):
        got = {}
        kwargs = {'device': device} if device else {}
        with make_batch_reader(url, shuffle_row_groups=False,
                               **kwargs) as r:
            for b in r:
                ids = np.asarray(b.id.cpu() if hasattr(b.id, 'cpu')
                                 else b.id)
                for i, rid in enumerate(ids):
                    row = {}
                    for f in b._fields:
                        v = getattr(b, f)
                        if hasattr(v, 'cpu'):
                            v = v.cpu().numpy()
                        row[f] = np.asarray(v)[i]
                    got[int(rid)] = row
            assist = list(r.diagnostics.get('cpu_assist_columns', []))
        return got, assist

    gpu, assist = read_all('cuda')
    cpu, _ = read_all(None)
    # scalar ints/floats/strings must be native on the GPU route
    for native_col in ('id', 'id_div_700', 'float64', 'string', 'string2'):
        assert native_col not in assist, (native_col, assist)
    assert len(gpu) == len(rows)
    for rid, crow in cpu.items():
        grow = gpu[rid]
        for f, cv in crow.items():
            gv = grow[f]
            if isinstance(cv, np.ndarray) or isinstance(gv, np.ndarray):
                np.testing.assert_array_equal(np.asarray(gv),
                                              np.asarray(cv), err_msg=f)
            else:
                assert gv == cv, (f, gv, cv)


# ---------------------------------------------------------------------------
# DELTA encodings (v2 writers: pyarrow column_encoding, Spark parquet v2)
# ---------------------------------------------------------------------------

def _write_delta_store(d, n=5000, compression='snappy', with_nulls=False):
    import pyarrow as pa
    import pyarrow.parquet as pq
    rng = np.random.RandomState(3)
    i64 = rng.randint(-2**40, 2**40, n).astype(np.int64)
    i64[::97] = np.iinfo(np.int64).max // 3   # big deltas
    i32 = rng.randint(-2**20, 2**20, n).astype(np.int32)
    strs = ['s%d-%s' % (i, 'x' * (i % 23)) for i in range(n)]
    svals = [None if (with_nulls and i % 11 == 0) else strs[i]
             for i in range(n)]
    table = pa.table({
        'i64': pa.array(i64),
        'i32': pa.array(i32),
        's': pa.array(svals, pa.string()),
    })
    pq.write_table(table, str(d / 'p.parquet'), row_group_size=1250,
                   use_dictionary=False, compression=compression,
                   column_encoding={'i64': 'DELTA_BINARY_PACKED',
                                    'i32': 'DELTA_BINARY_PACKED',
                                    's': 'DELTA_LENGTH_BYTE_ARRAY'},
                   data_page_size=16 << 10)
    return i64, i32, svals


@pytest.mark.parametrize('compression', ['snappy', 'none'])
def test_delta_encodings_gpu_exact(ext, tmp_path, compression):
    """DELTA_BINARY_PACKED ints + DELTA_LENGTH_BYTE_ARRAY strings decode
    on GPU, values exactly matching pyarrow."""
    from petastorm_amd import make_batch_reader
    d = tmp_path / ('delta_' + compression)
    d.mkdir()
    i64, i32, svals = _write_delta_store(d, compression=compression)
    got = {'i64': [], 'i32': [], 's': []}
    with make_batch_reader('file://' + str(d), device='cuda',
                           shuffle_row_groups=False) as r:
        for b in r:
            got['i64'].append(b.i64.cpu().numpy())
            got['i32'].append(b.i32.cpu().numpy())
            got['s'].append(np.asarray(b.s, dtype=object))
        assert r.diagnostics['cpu_assist_columns'] == []
    np.testing.assert_array_equal(np.concatenate(got['i64']), i64)
    np.testing.assert_array_equal(np.concatenate(got['i32']), i32)
    s_all = np.concatenate(got['s'])
    assert s_all.tolist() == svals


def test_delta_length_byte_array_with_nulls_gpu(ext, tmp_path):
    from petastorm_amd import make_batch_reader
    d = tmp_path / 'delta_nulls'
    d.mkdir()
    _, _, svals = _write_delta_store(d, n=2000, with_nulls=True)
    got = []
    with make_batch_reader('file://' + str(d), device='cuda',
                           shuffle_row_groups=False,
                           schema_fields=['s']) as r:
        for b in r:
            got.append(np.asarray(b.s, dtype=object))
        assert r.diagnostics['cpu_assist_columns'] == []
    assert np.concatenate(got).tolist() == svals


def test_bool_bss_delta_byte_array_gpu(ext, tmp_path):
    """BOOLEAN bit-unpack, BYTE_STREAM_SPLIT floats and DELTA_BYTE_ARRAY
    (front-coded) strings decode on GPU, exact vs source."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    from petastorm_amd import make_batch_reader
    d = tmp_path / 'enc3'
    d.mkdir()
    n = 4000
    rng = np.random.RandomState(5)
    f32 = rng.rand(n).astype(np.float32)
    f64 = rng.rand(n)
    bools = (np.arange(n) % 3 == 0)
    strs = ['shared-prefix-%06d/suffix-%s' % (i // 9, 'q' * (i % 13))
            for i in range(n)]
    table = pa.table({'id': pa.array(np.arange(n, dtype=np.int64)),
                      'f': pa.array(f32), 'd': pa.array(f64),
                      'b': pa.array(bools), 's': pa.array(strs)})
    pq.write_table(table, str(d / 'p.parquet'), row_group_size=1000,
                   use_dictionary=False, compression='snappy',
                   column_encoding={'f': 'BYTE_STREAM_SPLIT',
                                    'd': 'BYTE_STREAM_SPLIT',
                                    's': 'DELTA_BYTE_ARRAY',
                                    'id': 'PLAIN', 'b': 'PLAIN'},
                   data_page_size=8 << 10)
    got = {'id': [], 'f': [], 'd': [], 'b': [], 's': []}
    with make_batch_reader('file://' + str(d), device='cuda',
                           shuffle_row_groups=False) as r:
        for batch in r:
            for k in got:
                v = getattr(batch, k)
                got[k].append(v.cpu().numpy() if hasattr(v, 'cpu')
                              else np.asarray(v, dtype=object))
        assert r.diagnostics['cpu_assist_columns'] == []
    ids = np.concatenate(got['id'])
    np.testing.assert_array_equal(ids, np.arange(n))
    np.testing.assert_array_equal(np.concatenate(got['f']), f32)
    np.testing.assert_array_equal(np.concatenate(got['d']), f64)
    np.testing.assert_array_equal(np.concatenate(got['b']), bools)
    assert np.concatenate(got['s']).tolist() == strs


def test_int96_timestamps_gpu(ext, tmp_path):
    """Legacy Spark INT96 timestamps decode on GPU to datetime64[ns],
    matching the CPU route exactly."""
    import datetime
    import pyarrow as pa
    import pyarrow.parquet as pq
    from petastorm_amd import make_batch_reader
    d = tmp_path / 'int96'
    d.mkdir()
    n = 1000
    base = datetime.datetime(2001, 3, 4, 5, 6, 7)
    stamps = [base + datetime.timedelta(seconds=17 * i, microseconds=i % 997)
              for i in range(n)]
    table = pa.table({'id': pa.array(np.arange(n, dtype=np.int64)),
                      'ts': pa.array(stamps, pa.timestamp('us'))})
    pq.write_table(table, str(d / 'p.parquet'), row_group_size=250,
                   use_dictionary=False, compression='snappy',
                   use_deprecated_int96_timestamps=True)
    pf = pq.ParquetFile(str(d / 'p.parquet'))
    assert pf.metadata.row_group(0).column(1).physical_type == 'INT96'

    def read_all(device):
        got = {}
        kwargs = {'device': device} if device else {}
        with make_batch_reader('file://' + str(d), shuffle_row_groups=False,
                               **kwargs) as r:
            assists = None
            for b in r:
                ids = np.asarray(b.id.cpu() if hasattr(b.id, 'cpu')
                                 else b.id)
                ts = np.asarray(b.ts)
                for i, rid in enumerate(ids):
                    got[int(rid)] = ts[i]
            assists = list(r.diagnostics.get('cpu_assist_columns', []))
        return got, assists

    gpu, assist = read_all('cuda')
    cpu, _ = read_all(None)
    assert 'ts' not in assist
    for rid in range(n):
        assert np.datetime64(gpu[rid], 'ns') == np.datetime64(cpu[rid], 'ns')
        assert np.datetime64(gpu[rid], 'us') == np.datetime64(stamps[rid])


def test_flba_float16_and_fixed_binary_gpu(ext, tmp_path):
    """FIXED_LEN_BYTE_ARRAY: float16 columns view directly as half
    tensors; fixed-size binary surfaces bytes — exact values."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    from petastorm_amd import make_batch_reader
    d = tmp_path / 'flba'
    d.mkdir()
    n = 2000
    rng = np.random.RandomState(7)
    h = rng.rand(n).astype(np.float16)
    fb = [bytes([i % 251, (i * 7) % 251, 3, 4]) for i in range(n)]
    table = pa.table({'id': pa.array(np.arange(n, dtype=np.int64)),
                      'h': pa.array(h),
                      'fb': pa.array(fb, pa.binary(4))})
    pq.write_table(table, str(d / 'p.parquet'), row_group_size=500,
                   use_dictionary=False, compression='snappy')
    got_h, got_fb, got_id = [], [], []
    with make_batch_reader('file://' + str(d), device='cuda',
                           shuffle_row_groups=False) as r:
        for b in r:
            got_id.append(b.id.cpu().numpy())
            assert b.h.dtype == torch.float16
            got_h.append(b.h.cpu().numpy())
            got_fb.append(np.asarray(b.fb, dtype=object))
        assert r.diagnostics['cpu_assist_columns'] == []
    np.testing.assert_array_equal(np.concatenate(got_id), np.arange(n))
    np.testing.assert_array_equal(np.concatenate(got_h), h)
    assert np.concatenate(got_fb).tolist() == fb


def test_rle_boolean_column_gpu(ext, tmp_path):
    import pyarrow as pa
    import pyarrow.parquet as pq
    from petastorm_amd import make_batch_reader
    d = tmp_path / 'rlebool'
    d.mkdir()
    n = 3000
    vals = (np.arange(n) % 17 < 5) | (np.arange(n) % 97 == 0)
    table = pa.table({'id': pa.array(np.arange(n, dtype=np.int64)),
                      'b': pa.array(vals)})
    pq.write_table(table, str(d / 'p.parquet'), row_group_size=750,
                   use_dictionary=False, compression='snappy',
                   column_encoding={'b': 'RLE', 'id': 'PLAIN'})
    got_b, got_id = [], []
    with make_batch_reader('file://' + str(d), device='cuda',
                           shuffle_row_groups=False) as r:
        for batch in r:
            got_id.append(batch.id.cpu().numpy())
            got_b.append(batch.b.cpu().numpy())
        assert r.diagnostics['cpu_assist_columns'] == []
    np.testing.assert_array_equal(np.concatenate(got_id), np.arange(n))
    np.testing.assert_array_equal(np.concatenate(got_b), vals)


def test_batched_dataloader_over_gpu_reader(ext, tmp_path):
    """BatchedDataLoader + on-device shuffling pool over the GPU reader:
    all rows delivered once per epoch, tensors stay on device."""
    from petastorm_amd import make_batch_reader
    from petastorm_amd.pytorch import BatchedDataLoader
    from petastorm_amd.test_util.dataset_gen import create_scalar_dataset
    url = 'file://' + str(tmp_path / 'ld')
    create_scalar_dataset(url, num_rows=3000, rowgroup_size=500)
    reader = make_batch_reader(url, device='cuda', num_epochs=1, seed=11,
                               shuffle_row_groups=True,
                               schema_fields=['id', 'f0'])
    loader = BatchedDataLoader(reader, batch_size=256,
                               shuffling_queue_capacity=1024, seed=5)
    ids = []
    for b in loader:
        assert b['id'].is_cuda
        ids.extend(b['id'].cpu().numpy().tolist())
    reader.stop()
    reader.join()
    assert sorted(ids) == list(range(3000))
    assert ids != sorted(ids)  # the pool actually shuffled


def test_inmem_batched_dataloader_gpu(ext, tmp_path):
    from petastorm_amd import make_batch_reader
    from petastorm_amd.pytorch import InMemBatchedDataLoader
    from petastorm_amd.test_util.dataset_gen import create_scalar_dataset
    url = 'file://' + str(tmp_path / 'inmem')
    create_scalar_dataset(url, num_rows=1000, rowgroup_size=250)
    reader = make_batch_reader(url, device='cuda', num_epochs=1,
                               shuffle_row_groups=False,
                               schema_fields=['id', 'i0'])
    loader = InMemBatchedDataLoader(reader, batch_size=128, num_epochs=2,
                                    rows_capacity=1000, shuffle=True,
                                    seed=3)
    epochs_ids = []
    for _ in range(2):  # one iter() per epoch (reference pytorch.py:437+)
        ids = []
        for b in loader:
            ids.extend(b['id'].cpu().numpy().tolist())
        epochs_ids.append(ids)
    reader.stop()
    reader.join()
    assert len(epochs_ids) == 2
    assert sorted(epochs_ids[0]) == list(range(1000))
    assert sorted(epochs_ids[1]) == list(range(1000))
    assert epochs_ids[0] != epochs_ids[1]  # per-epoch reshuffle


def test_gpu_reader_reset_and_string_predicate(ext, tmp_path):
    """reset() restarts epoch iteration on the GPU route, and predicates
    over STRING fields (host-evaluated mask on device columns) filter
    correctly — incl. the upstream-exact pseudorandom split."""
    from petastorm_amd import make_batch_reader
    from petastorm_amd.predicates import in_lambda, in_pseudorandom_split
    from petastorm_amd.test_util.dataset_gen import create_scalar_dataset
    url = 'file://' + str(tmp_path / 'resetp')
    cols = create_scalar_dataset(url, num_rows=1000, rowgroup_size=250)

    pred = in_lambda(['name'], lambda v: np.char.endswith(
        np.asarray(v['name'], dtype=np.str_), '7'))
    with make_batch_reader(url, device='cuda', num_epochs=1,
                           shuffle_row_groups=False, predicate=pred,
                           schema_fields=['id', 'name']) as r:
        ids1 = np.sort(np.concatenate(
            [b.id.cpu().numpy() for b in r]))
        r.reset()
        ids2 = np.sort(np.concatenate(
            [b.id.cpu().numpy() for b in r]))
    expected = np.array([i for i in range(1000)
                         if cols['name'][i].endswith('7')])
    np.testing.assert_array_equal(ids1, expected)
    np.testing.assert_array_equal(ids2, expected)

    split = in_pseudorandom_split([0.6, 0.4], 0, 'name')
    with make_batch_reader(url, device='cuda', num_epochs=1,
                           shuffle_row_groups=False, predicate=split,
                           schema_fields=['id', 'name']) as r:
        got = np.sort(np.concatenate([b.id.cpu().numpy() for b in r]))
    keep = np.array([i for i in range(1000)
                     if split.do_include({'name': cols['name'][i]})])
    np.testing.assert_array_equal(got, keep)


def test_v2_pages_delta_and_bool_gpu(ext, tmp_path):
    """DataPageV2 with the widened encodings (delta ints, delta-length
    strings, byte-stream-split floats, plain bools): exact values,
    snappy-compressed values sections."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    from petastorm_amd import make_batch_reader
    d = tmp_path / 'v2enc'
    d.mkdir()
    n = 3000
    rng = np.random.RandomState(9)
    i64 = rng.randint(-2**50, 2**50, n)
    f32 = rng.rand(n).astype(np.float32)
    strs = ['v2-%d-%s' % (i, 'z' * (i % 11)) for i in range(n)]
    bools = np.arange(n) % 5 < 2
    table = pa.table({'id': pa.array(np.arange(n, dtype=np.int64)),
                      'i64': pa.array(i64), 'f': pa.array(f32),
                      's': pa.array(strs), 'b': pa.array(bools)})
    pq.write_table(table, str(d / 'p.parquet'), row_group_size=750,
                   use_dictionary=False, compression='snappy',
                   data_page_version='2.0', data_page_size=16 << 10,
                   column_encoding={'i64': 'DELTA_BINARY_PACKED',
                                    'f': 'BYTE_STREAM_SPLIT',
                                    's': 'DELTA_LENGTH_BYTE_ARRAY',
                                    'id': 'PLAIN', 'b': 'PLAIN'})
    got = {k: [] for k in ('id', 'i64', 'f', 's', 'b')}
    with make_batch_reader('file://' + str(d), device='cuda',
                           shuffle_row_groups=False) as r:
        for batch in r:
            for k in got:
                v = getattr(batch, k)
                got[k].append(v.cpu().numpy() if hasattr(v, 'cpu')
                              else np.asarray(v, dtype=object))
        assert r.diagnostics['cpu_assist_columns'] == []
    np.testing.assert_array_equal(np.concatenate(got['id']), np.arange(n))
    np.testing.assert_array_equal(np.concatenate(got['i64']), i64)
    np.testing.assert_array_equal(np.concatenate(got['f']), f32)
    np.testing.assert_array_equal(np.concatenate(got['b']), bools)
    assert np.concatenate(got['s']).tolist() == strs


def test_unsigned_logical_types_gpu(ext, tmp_path):
    """Unsigned logical types over signed physical storage: values exact
    at full range (uint8/16/32/64), widened per the framework convention
    (u16->i32, u32->i64, u64->numpy)."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    from petastorm_amd import make_batch_reader
    d = tmp_path / 'uns'
    d.mkdir()
    n = 300
    rng = np.random.RandomState(13)
    u8 = rng.randint(0, 256, n).astype(np.uint8)
    u16 = rng.randint(0, 2**16, n).astype(np.uint16)
    u32 = (rng.randint(0, 2**31, n).astype(np.uint64) * 2 + 1) \
        .astype(np.uint32)
    u64 = (rng.randint(0, 2**62, n).astype(np.uint64) * 4 + 3)
    table = pa.table({'id': pa.array(np.arange(n, dtype=np.int64)),
                      'u8': pa.array(u8), 'u16': pa.array(u16),
                      'u32': pa.array(u32), 'u64': pa.array(u64)})
    pq.write_table(table, str(d / 'p.parquet'), row_group_size=100,
                   use_dictionary=False, compression='snappy')
    got = {k: [] for k in ('id', 'u8', 'u16', 'u32', 'u64')}
    with make_batch_reader('file://' + str(d), device='cuda',
                           shuffle_row_groups=False) as r:
        for b in r:
            assert b.u8.dtype == torch.uint8
            assert b.u16.dtype == torch.int32
            assert b.u32.dtype == torch.int64
            assert isinstance(b.u64, np.ndarray) and \
                b.u64.dtype == np.uint64
            for k in got:
                v = getattr(b, k)
                got[k].append(v.cpu().numpy() if hasattr(v, 'cpu')
                              else np.asarray(v))
        assert r.diagnostics['cpu_assist_columns'] == []
    np.testing.assert_array_equal(np.concatenate(got['u8']), u8)
    np.testing.assert_array_equal(np.concatenate(got['u16']),
                                  u16.astype(np.int32))
    np.testing.assert_array_equal(np.concatenate(got['u32']),
                                  u32.astype(np.int64))
    np.testing.assert_array_equal(np.concatenate(got['u64']), u64)


def test_v2_zstd_delta_gpu(ext, tmp_path):
    """V2 pages with ZSTD values sections + delta encodings: the host
    zstd stage rebases pages, the widened V2 dispatch decodes."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    from petastorm_amd import make_batch_reader
    d = tmp_path / 'v2zstd'
    d.mkdir()
    n = 2000
    i64 = np.cumsum(np.random.RandomState(1).randint(-50, 50, n)) \
        .astype(np.int64)
    strs = ['zz-%04d' % (i % 137) for i in range(n)]
    table = pa.table({'id': pa.array(np.arange(n, dtype=np.int64)),
                      'i64': pa.array(i64), 's': pa.array(strs)})
    pq.write_table(table, str(d / 'p.parquet'), row_group_size=500,
                   use_dictionary=False, compression='zstd',
                   data_page_version='2.0',
                   column_encoding={'i64': 'DELTA_BINARY_PACKED',
                                    's': 'DELTA_LENGTH_BYTE_ARRAY',
                                    'id': 'PLAIN'})
    got = {'id': [], 'i64': [], 's': []}
    with make_batch_reader('file://' + str(d), device='cuda',
                           shuffle_row_groups=False) as r:
        for b in r:
            got['id'].append(b.id.cpu().numpy())
            got['i64'].append(b.i64.cpu().numpy())
            got['s'].append(np.asarray(b.s, dtype=object))
        assert r.diagnostics['cpu_assist_columns'] == []
    np.testing.assert_array_equal(np.concatenate(got['id']), np.arange(n))
    np.testing.assert_array_equal(np.concatenate(got['i64']), i64)
    assert np.concatenate(got['s']).tolist() == strs


def test_corrupt_jpeg_fails_loudly_gpu(ext, tmp_path):
    """A corrupted jpeg payload must raise a decode error (status
    machinery), never hang or return garbage silently."""
    import glob
    import pyarrow.parquet as pq
    import pyarrow as pa
    from petastorm_amd import make_batch_reader
    from petastorm_amd.test_util.dataset_gen import create_imagenet_dataset
    url_dir = str(tmp_path / 'corrupt')
    create_imagenet_dataset('file://' + url_dir, num_rows=8,
                            rowgroup_size_mb=8)
    f = glob.glob(url_dir + '/*.parquet')[0]
    t = pq.read_table(f)
    imgs = t.column('image').to_pylist()
    bad = bytearray(imgs[3])
    # corrupt the header structure (invalid marker lengths).  NB: stomping
    # ENTROPY bytes instead hits the graceful truncated-stream path — an
    # unstuffed 0xFF reads as end-of-segment and the remaining MCUs decode
    # as zero-padded data, libjpeg's incomplete-stream behavior.
    sos = bytes(bad).find(b'\xff\xda')
    assert sos > 0
    bad[2:sos] = bytes(len(bad[2:sos]))  # zero every header segment
    imgs[3] = bytes(bad)
    t2 = t.set_column(t.schema.get_field_index('image'), 'image',
                      pa.array(imgs, pa.binary()))
    pq.write_table(t2, f, row_group_size=8, use_dictionary=False,
                   compression='none')
    with pytest.raises(Exception) as ei:
        with make_batch_reader('file://' + url_dir, device='cuda',
                               shuffle_row_groups=False) as r:
            list(r)
    msg = str(ei.value).lower()
    assert 'jpeg' in msg or 'decode' in msg or 'marker' in msg or \
        'image' in msg


def test_two_concurrent_gpu_readers(ext, tmp_path):
    """Two independent GPU readers in one process (separate stream pools)
    interleave without interference."""
    from petastorm_amd import make_batch_reader
    from petastorm_amd.test_util.dataset_gen import create_scalar_dataset
    u1 = 'file://' + str(tmp_path / 'a')
    u2 = 'file://' + str(tmp_path / 'b')
    create_scalar_dataset(u1, num_rows=2000, rowgroup_size=500, seed=1)
    create_scalar_dataset(u2, num_rows=1500, rowgroup_size=500, seed=2)
    r1 = make_batch_reader(u1, device='cuda', num_epochs=1,
                           shuffle_row_groups=False, schema_fields=['id'])
    r2 = make_batch_reader(u2, device='cuda', num_epochs=1,
                           shuffle_row_groups=False, schema_fields=['id'])
    it1, it2 = iter(r1), iter(r2)
    got1, got2 = [], []
    while True:
        done = 0
        for it, acc in ((it1, got1), (it2, got2)):
            try:
                acc.append(next(it).id.cpu().numpy())
            except StopIteration:
                done += 1
        if done == 2:
            break
    for r in (r1, r2):
        r.stop()
        r.join()
    assert sorted(np.concatenate(got1).tolist()) == list(range(2000))
    assert sorted(np.concatenate(got2).tolist()) == list(range(1500))


def test_throughput_harness_gpu(ext, tmp_path):
    """The reference-shaped benchmark harness runs against the GPU batch
    reader (reference throughput.py:112-172 with device='cuda')."""
    from petastorm_amd.benchmark.throughput import reader_throughput
    from petastorm_amd.test_util.dataset_gen import create_scalar_dataset
    url = 'file://' + str(tmp_path / 'thr')
    create_scalar_dataset(url, num_rows=5000, rowgroup_size=1000)
    res = reader_throughput(url, warmup_cycles_count=2,
                            measure_cycles_count=5, read_method='batch',
                            device='cuda')
    assert res.samples_per_second > 0


def test_corrupt_delta_page_fails_loudly_gpu(ext, tmp_path):
    """Garbage bytes in a DELTA_LENGTH_BYTE_ARRAY page must raise, never
    gather out of bounds or hang."""
    import glob
    import pyarrow as pa
    import pyarrow.parquet as pq
    from petastorm_amd import make_batch_reader
    d = tmp_path / 'cdelta'
    d.mkdir()
    n = 500
    table = pa.table({'s': pa.array(['x%04d' % i for i in range(n)])})
    pq.write_table(table, str(d / 'p.parquet'), use_dictionary=False,
                   compression='none',
                   column_encoding={'s': 'DELTA_LENGTH_BYTE_ARRAY'})
    f = glob.glob(str(d) + '/*.parquet')[0]
    raw = bytearray(open(f, 'rb').read())
    pf = pq.ParquetFile(f)
    col = pf.metadata.row_group(0).column(0)
    off = col.data_page_offset
    # stomp the delta header + first miniblocks with 0xAA garbage
    for i in range(off + 12, off + 200):
        raw[i] = 0xAA
    open(f, 'wb').write(bytes(raw))
    with pytest.raises(Exception):
        with make_batch_reader('file://' + str(d), device='cuda',
                               shuffle_row_groups=False) as r:
            list(r)


def test_int96_nullable_nat_gpu(ext, tmp_path):
    """Nullable INT96 with actual nulls: null rows surface as NaT (CPU
    route parity)."""
    import datetime
    import pyarrow as pa
    import pyarrow.parquet as pq
    from petastorm_amd import make_batch_reader
    d = tmp_path / 'int96n'
    d.mkdir()
    n = 200
    base = datetime.datetime(2015, 6, 1)
    stamps = [None if i % 7 == 0 else
              base + datetime.timedelta(minutes=i) for i in range(n)]
    table = pa.table({'id': pa.array(np.arange(n, dtype=np.int64)),
                      'ts': pa.array(stamps, pa.timestamp('us'))})
    pq.write_table(table, str(d / 'p.parquet'), row_group_size=50,
                   use_dictionary=False, compression='snappy',
                   use_deprecated_int96_timestamps=True)
    got = {}
    with make_batch_reader('file://' + str(d), device='cuda',
                           shuffle_row_groups=False) as r:
        for b in r:
            ids = b.id.cpu().numpy()
            ts = np.asarray(b.ts)
            for i, rid in enumerate(ids):
                got[int(rid)] = ts[i]
        assert 'ts' not in r.diagnostics['cpu_assist_columns']
    for i in range(n):
        if stamps[i] is None:
            assert np.isnat(got[i]), (i, got[i])
        else:
            assert np.datetime64(got[i], 'us') == np.datetime64(stamps[i])


def test_nullable_datetime_nat_gpu(ext, tmp_path):
    """Nullable DATE/TIMESTAMP (INT32/INT64 physical) with actual nulls:
    NaT on the GPU route (sentinel fill + boundary conversion)."""
    import datetime
    import pyarrow as pa
    import pyarrow.parquet as pq
    from petastorm_amd import make_batch_reader
    d = tmp_path / 'dtnull'
    d.mkdir()
    n = 120
    dates = [None if i % 5 == 0 else
             datetime.date(2020, 1, 1) + datetime.timedelta(days=i)
             for i in range(n)]
    stamps = [None if i % 4 == 0 else
              datetime.datetime(2021, 2, 3) + datetime.timedelta(seconds=i)
              for i in range(n)]
    table = pa.table({'id': pa.array(np.arange(n, dtype=np.int64)),
                      'd': pa.array(dates, pa.date32()),
                      'ts': pa.array(stamps, pa.timestamp('us'))})
    pq.write_table(table, str(d / 'p.parquet'), row_group_size=40,
                   use_dictionary=False, compression='snappy')
    got = {}
    with make_batch_reader('file://' + str(d), device='cuda',
                           shuffle_row_groups=False) as r:
        for b in r:
            ids = b.id.cpu().numpy()
            for i, rid in enumerate(ids):
                got[int(rid)] = (np.asarray(b.d)[i], np.asarray(b.ts)[i])
        assert r.diagnostics['cpu_assist_columns'] == []
    for i in range(n):
        gd, gts = got[i]
        if dates[i] is None:
            assert np.isnat(gd)
        else:
            assert np.datetime64(gd, 'D') == np.datetime64(dates[i], 'D')
        if stamps[i] is None:
            assert np.isnat(gts)
        else:
            assert np.datetime64(gts, 'us') == np.datetime64(stamps[i])
