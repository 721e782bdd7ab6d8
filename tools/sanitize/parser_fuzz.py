"""Heavy host-parser fuzz harness (the in-suite test runs a 150-trial
subset; this is the full sweep run during development — all clean as of
r2.9 after the bounds fixes it motivated; see profiles/RESULTS.md).

    python tools/sanitize/parser_fuzz.py [--trials N]

Covers jpeg_parse_batch (4:4:4/4:2:2/4:2:0, grayscale, restart markers),
png_parse_batch (rgb/gray/rgba/16-bit) and parquet_walk_pages
(V1/V2/dict/DELTA stores): N bit-flip trials per variant plus every
truncation point.  A crash (not a RuntimeError) is a bug.
"""
import argparse
import glob  # noqa: F401
import io
import os
import sys
import tempfile

import numpy as np
import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))


def jpeg_png_corpus(rng):
    from PIL import Image
    from petastorm_amd import ops
    e = ops.ext()
    corpus = []
    img = Image.fromarray(rng.randint(0, 255, (40, 56, 3)).astype(np.uint8))
    for ss in (0, 1, 2):
        b = io.BytesIO()
        img.save(b, format='JPEG', quality=92, subsampling=ss)
        corpus.append(('jpeg-ss%d' % ss, e.jpeg_parse_batch, b.getvalue()))
    g = Image.fromarray(rng.randint(0, 255, (48, 48)).astype(np.uint8))
    b = io.BytesIO(); g.save(b, format='JPEG', quality=60)
    corpus.append(('jpeg-gray', e.jpeg_parse_batch, b.getvalue()))
    b = io.BytesIO()
    img.save(b, format='JPEG', quality=92, restart_marker_rows=1)
    corpus.append(('jpeg-rst', e.jpeg_parse_batch, b.getvalue()))
    for tag, im in (('png-rgb', img), ('png-gray', g)):
        b = io.BytesIO(); im.save(b, format='PNG')
        corpus.append((tag, e.png_parse_batch, b.getvalue()))
    ra = Image.fromarray(rng.randint(0, 255, (32, 32, 4)).astype(np.uint8),
                         'RGBA')
    b = io.BytesIO(); ra.save(b, format='PNG')
    corpus.append(('png-rgba', e.png_parse_batch, b.getvalue()))
    g16 = Image.fromarray(rng.randint(0, 65535, (24, 24)).astype(np.uint16),
                          'I;16')
    b = io.BytesIO(); g16.save(b, format='PNG')
    corpus.append(('png-16', e.png_parse_batch, b.getvalue()))
    return corpus


def fuzz_blob(name, parse, blob, rng, trials):
    for _ in range(trials):
        buf = bytearray(blob)
        for _ in range(rng.randint(1, 9)):
            buf[rng.randint(0, len(buf))] ^= 1 << rng.randint(0, 8)
        t = torch.frombuffer(bytes(buf), dtype=torch.uint8)
        try:
            parse(t, torch.tensor([0]), torch.tensor([len(buf)]))
        except RuntimeError:
            pass
    for cut in range(1, len(blob)):
        t = torch.frombuffer(bytes(blob[:cut]), dtype=torch.uint8)
        try:
            parse(t, torch.tensor([0]), torch.tensor([cut]))
        except RuntimeError:
            pass
    print(name, 'ok', flush=True)


def fuzz_thrift(rng, trials):
    import pyarrow as pa
    import pyarrow.parquet as pq
    from petastorm_amd import ops
    e = ops.ext()

    def store(tag, **kw):
        d = tempfile.mkdtemp(prefix='fz_' + tag)
        t = pa.table({'a': pa.array(rng.randint(0, 1000, 5000,
                                                dtype=np.int64)),
                      's': pa.array(['v%d' % i for i in range(5000)])})
        pq.write_table(t, d + '/f.parquet', row_group_size=1000, **kw)
        return tag, d + '/f.parquet'

    stores = [
        store('v1', use_dictionary=False),
        store('v2', use_dictionary=False, data_page_version='2.0'),
        store('dict', use_dictionary=True),
        store('delta', use_dictionary=False,
              column_encoding={'a': 'DELTA_BINARY_PACKED',
                               's': 'DELTA_LENGTH_BYTE_ARRAY'}),
    ]
    for tag, f in stores:
        raw = bytearray(open(f, 'rb').read())
        md = pq.ParquetFile(f).metadata
        col = md.row_group(0).column(0)
        start = col.data_page_offset
        if col.dictionary_page_offset is not None:
            start = min(start, col.dictionary_page_offset)
        span = int(col.total_compressed_size)
        for _ in range(trials):
            buf = bytearray(raw)
            for _ in range(rng.randint(1, 6)):
                pos = start + rng.randint(0, span)
                buf[pos] ^= 1 << rng.randint(0, 8)
            host = torch.frombuffer(bytes(buf), dtype=torch.uint8)
            try:
                e.parquet_walk_pages(host, torch.tensor([start]),
                                     torch.tensor([span]))
            except RuntimeError:
                pass
        print('thrift-' + tag, 'ok', flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--trials', type=int, default=4000)
    ap.add_argument('--seed', type=int, default=2024)
    args = ap.parse_args()
    rng = np.random.RandomState(args.seed)
    for name, parse, blob in jpeg_png_corpus(rng):
        fuzz_blob(name, parse, blob, rng, args.trials)
    fuzz_thrift(rng, args.trials)
    print('ALL OK')


if __name__ == '__main__':
    main()
