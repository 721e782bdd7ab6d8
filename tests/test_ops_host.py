"""Host-side pieces of the HIP extension (run on CPU-only machines):
thrift page walker and the JPEG header parser."""
import io

import numpy as np
import pytest
import torch

from petastorm_amd import ops

pytestmark = pytest.mark.skipif(not ops.available(),
                                reason='HIP extension not built')


def _ext():
    return ops.ext()


def test_parquet_walk_pages_matches_pyarrow(scalar_dataset):
    import pyarrow.parquet as pq
    e = _ext()
    import glob
    f = sorted(glob.glob(scalar_dataset['path'] + '/*.parquet'))[0]
    raw = open(f, 'rb').read()
    host = torch.frombuffer(bytearray(raw), dtype=torch.uint8)
    md = pq.ParquetFile(f).metadata
    for rg in range(md.num_row_groups):
        offs, lens, nvals = [], [], []
        for ci in range(md.num_columns):
            col = md.row_group(rg).column(ci)
            start = col.data_page_offset
            if col.dictionary_page_offset is not None:
                start = min(start, col.dictionary_page_offset)
            offs.append(start)
            lens.append(col.total_compressed_size)
            nvals.append(col.num_values)
        pages = e.parquet_walk_pages(host, torch.tensor(offs),
                                     torch.tensor(lens))
        for ci in range(md.num_columns):
            got = int(pages['num_values'][
                (pages['page_chunk'] == ci) & (pages['page_type'] == 0)].sum())
            assert got == nvals[ci]


def _make_jpegs(n, size=(48, 64), quality=90, gray=False, subsampling=None):
    from PIL import Image
    rng = np.random.RandomState(0)
    blobs, arrays = [], []
    for i in range(n):
        if gray:
            arr = rng.randint(0, 255, size[::-1]).astype(np.uint8)
        else:
            arr = rng.randint(0, 255, size[::-1] + (3,)).astype(np.uint8)
        img = Image.fromarray(arr)
        b = io.BytesIO()
        kw = dict(format='JPEG', quality=quality, restart_marker_rows=1)
        if subsampling is not None:
            kw['subsampling'] = subsampling
        img.save(b, **kw)
        blobs.append(b.getvalue())
        arrays.append(arr)
    return blobs, arrays


def test_jpeg_parse_batch_geometry():
    e = _ext()
    blobs, _ = _make_jpegs(4)
    buf = b''.join(blobs)
    off, lens, pos = [], [], 0
    for d in blobs:
        off.append(pos)
        lens.append(len(d))
        pos += len(d)
    t = torch.frombuffer(bytearray(buf), dtype=torch.uint8)
    meta = e.jpeg_parse_batch(t, torch.tensor(off), torch.tensor(lens))
    assert meta['width'].tolist() == [48] * 4
    assert meta['height'].tolist() == [64] * 4
    assert meta['ncomp'].tolist() == [3] * 4
    # 4:2:0 -> mcu 16x16 -> 3x4 grid; one restart segment per MCU row
    assert meta['mcus_x'].tolist() == [3] * 4
    assert meta['mcus_y'].tolist() == [4] * 4
    assert meta['seg_img'].numel() == 16
    assert int(meta['seg_nmcu'].sum()) == 4 * 12


def test_jpeg_parse_rejects_progressive():
    from PIL import Image
    e = _ext()
    rng = np.random.RandomState(0)
    img = Image.fromarray(rng.randint(0, 255, (32, 32, 3)).astype(np.uint8))
    b = io.BytesIO()
    img.save(b, format='JPEG', progressive=True)
    d = b.getvalue()
    t = torch.frombuffer(bytearray(d), dtype=torch.uint8)
    with pytest.raises(RuntimeError, match='non-baseline'):
        e.jpeg_parse_batch(t, torch.tensor([0]), torch.tensor([len(d)]))


def test_walk_pages_corrupt_binary_length_fails_loudly():
    """A corrupt T_BINARY length >= 2^63 casts to a negative skip; the
    cursor must fail (raise) instead of moving backwards and re-parsing
    forever (thrift_pages.cpp skip_bytes)."""
    e = _ext()
    # Thrift compact struct: field delta=4 (crc, skipped via skip_value)
    # declared as T_BINARY(8), then a 10-byte uvarint >= 2^63.
    blob = bytes([0x48]) + bytes([0xFF] * 9) + bytes([0x01]) + bytes(16)
    t = torch.frombuffer(bytearray(blob), dtype=torch.uint8)
    with pytest.raises(RuntimeError):
        e.parquet_walk_pages(t, torch.tensor([0]),
                             torch.tensor([len(blob)]))


def test_zstd_decoder_under_asan_ubsan():
    """Build and run the standalone ASAN/UBSan fuzz harness over the
    from-scratch zstd decoder (sanitizer coverage for host
    code).  Malformed input must produce error codes, never an
    out-of-bounds access."""
    import os
    import subprocess
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    src = os.path.join(root, 'tools', 'sanitize', 'zstd_fuzz_main.cpp')
    csrc = os.path.join(root, 'petastorm_amd', 'ops', 'csrc')
    exe = os.path.join(root, 'tools', 'sanitize', '.zstd_fuzz_bin')
    build = subprocess.run(
        ['g++', '-std=c++17', '-O1', '-g',
         '-fsanitize=address,undefined', '-fno-sanitize-recover=all',
         '-I', csrc, src, '-o', exe],
        capture_output=True, text=True, timeout=300)
    assert build.returncode == 0, build.stderr[-2000:]
    run = subprocess.run([exe], capture_output=True, text=True,
                         timeout=300)
    assert run.returncode == 0, (run.stdout + run.stderr)[-2000:]
    assert 'zstd fuzz OK' in run.stdout


def test_walk_pages_robust_to_garbage(scalar_dataset):
    """The thrift walker must raise cleanly (never crash) on arbitrary
    bytes and on bit-flipped real column chunks."""
    import glob
    e = _ext()
    rng = np.random.RandomState(7)
    # pure garbage buffers of varied sizes
    for n in (0, 1, 2, 7, 64, 4096):
        junk = torch.from_numpy(rng.randint(0, 256, n, dtype=np.uint8))
        try:
            e.parquet_walk_pages(junk, torch.tensor([0]),
                                 torch.tensor([n]))
        except RuntimeError:
            pass  # clean rejection is the contract
    # bit-flipped real chunk: every outcome must be raise-or-return
    import pyarrow.parquet as pq
    f = sorted(glob.glob(scalar_dataset['path'] + '/*.parquet'))[0]
    raw = bytearray(open(f, 'rb').read())
    md = pq.ParquetFile(f).metadata
    col = md.row_group(0).column(0)
    start = col.data_page_offset
    if col.dictionary_page_offset is not None:
        start = min(start, col.dictionary_page_offset)
    for trial in range(200):
        buf = bytearray(raw)
        for _ in range(rng.randint(1, 4)):
            pos = start + rng.randint(0, max(1, min(64, len(raw) - start)))
            buf[pos] ^= 1 << rng.randint(0, 8)
        host = torch.frombuffer(bytes(buf), dtype=torch.uint8)
        try:
            e.parquet_walk_pages(host, torch.tensor([start]),
                                 torch.tensor([col.total_compressed_size]))
        except RuntimeError:
            pass


def test_image_parsers_robust_to_corruption():
    """jpeg/png host parsers must raise or return on bit-flipped streams,
    never crash the process."""
    from PIL import Image
    e = _ext()
    rng = np.random.RandomState(11)
    img = Image.fromarray(rng.randint(0, 255, (32, 48, 3)).astype(np.uint8))
    bj = io.BytesIO(); img.save(bj, format='JPEG', quality=85)
    bp = io.BytesIO(); img.save(bp, format='PNG')
    for parse, blob in ((e.jpeg_parse_batch, bj.getvalue()),
                        (e.png_parse_batch, bp.getvalue())):
        for trial in range(150):
            buf = bytearray(blob)
            for _ in range(rng.randint(1, 5)):
                buf[rng.randint(0, len(buf))] ^= 1 << rng.randint(0, 8)
            t = torch.frombuffer(bytes(buf), dtype=torch.uint8)
            try:
                parse(t, torch.tensor([0]), torch.tensor([len(buf)]))
            except RuntimeError:
                pass
        # truncations at every prefix of the header region
        for cut in range(1, min(64, len(blob))):
            t = torch.frombuffer(bytes(blob[:cut]), dtype=torch.uint8)
            try:
                parse(t, torch.tensor([0]), torch.tensor([cut]))
            except RuntimeError:
                pass


def test_byte_array_host_offsets_rejects_overrun():
    """Corrupt varlen length prefixes must raise, not walk off the buffer."""
    e = _ext()
    # two values: 3 bytes, then a length claiming 2^31
    import struct
    payload = struct.pack('<I', 3) + b'abc' + struct.pack('<I', 0x7fffffff)
    t = torch.frombuffer(bytearray(payload + b'xy'), dtype=torch.uint8)
    with pytest.raises(RuntimeError, match='overruns'):
        e.byte_array_host_offsets(t, torch.tensor([0]), torch.tensor([2]))
    # truncated length prefix
    t2 = torch.frombuffer(bytearray(b'\x03\x00'), dtype=torch.uint8)
    with pytest.raises(RuntimeError, match='overruns'):
        e.byte_array_host_offsets(t2, torch.tensor([0]), torch.tensor([1]))
    # valid input still works
    ok = struct.pack('<I', 3) + b'abc' + struct.pack('<I', 1) + b'z'
    t3 = torch.frombuffer(bytearray(ok), dtype=torch.uint8)
    res = e.byte_array_host_offsets(t3, torch.tensor([0]), torch.tensor([2]))
    assert res['off'].tolist() == [4, 11] and res['len'].tolist() == [3, 1]
