"""Re-encode an imported dataset's JPEG columns with restart markers.

Foreign baseline JPEGs (cv2/PIL defaults — including everything upstream
petastorm ever wrote through CompressedImageCodec, reference codecs.py:97)
carry no RSTn markers, so the MI355X Huffman kernel decodes them as ONE
segment per image: correct, but the chip runs underfilled (a 224px image
is ~784 sequential MCUs for a single thread).  This tool rewrites such a
dataset once, re-encoding every jpeg CompressedImageCodec column with an
RSTn every ``--rst-blocks`` MCUs (the sweep-measured optimum is 2,
profiles/RESULTS.md), after which the GPU route decodes each image as
~100 independent segments.

Lossy note: a JPEG re-encode is a decode+encode round trip (one extra
generation loss at ``--quality``).  For lossless import keep the original
dataset and accept single-segment GPU decode or the CPU route.

Works on this framework's datasets AND on upstream-petastorm stores (the
pickled-metadata interop reader supplies the schema).
"""

import argparse
import os
import sys

from petastorm_amd import make_reader
from petastorm_amd.codecs import CompressedImageCodec
from petastorm_amd.etl.dataset_metadata import materialize_dataset
from petastorm_amd.unischema import Unischema, UnischemaField


def reencode_dataset(source_url, target_url, rst_blocks=2, quality=90,
                     rowgroup_size_mb=32, compression='snappy'):
    """Stream-copy ``source_url`` to ``target_url``, re-encoding every
    jpeg-codec column with restart markers.  Returns (rows, jpeg_columns)."""
    prev = os.environ.get('PSA_JPEG_RST_BLOCKS')
    os.environ['PSA_JPEG_RST_BLOCKS'] = str(rst_blocks)
    try:
        with make_reader(source_url, shuffle_row_groups=False,
                         reader_pool_type='thread',
                         workers_count=4) as reader:
            fields = []
            jpeg_cols = []
            for f in reader.schema.fields.values():
                if isinstance(f.codec, CompressedImageCodec) and \
                        f.codec.image_codec == 'jpeg':
                    jpeg_cols.append(f.name)
                    fields.append(UnischemaField(
                        f.name, f.numpy_dtype, f.shape,
                        CompressedImageCodec('jpeg', quality=quality),
                        f.nullable))
                else:
                    fields.append(f)
            if not jpeg_cols:
                raise ValueError(
                    'Dataset at {} has no jpeg CompressedImageCodec '
                    'columns to re-encode'.format(source_url))
            out_schema = Unischema('reencoded', fields)
            rows = 0
            with materialize_dataset(target_url, out_schema,
                                     rowgroup_size_mb,
                                     compression) as writer:
                for row in reader:
                    writer.write_row(row._asdict())
                    rows += 1
        return rows, jpeg_cols
    finally:
        if prev is None:
            os.environ.pop('PSA_JPEG_RST_BLOCKS', None)
        else:
            os.environ['PSA_JPEG_RST_BLOCKS'] = prev


def main(args=None):
    ap = argparse.ArgumentParser(
        description='Re-encode jpeg columns with restart markers for '
                    'parallel GPU Huffman decode')
    ap.add_argument('source_url')
    ap.add_argument('target_url')
    ap.add_argument('--rst-blocks', type=int, default=2,
                    help='MCUs per restart segment (default 2, the '
                         'MI355X sweep optimum)')
    ap.add_argument('--quality', type=int, default=90)
    ap.add_argument('--rowgroup-size-mb', type=int, default=32)
    ap.add_argument('--compression', default='snappy')
    a = ap.parse_args(args)
    rows, cols = reencode_dataset(a.source_url, a.target_url, a.rst_blocks,
                                  a.quality, a.rowgroup_size_mb,
                                  a.compression)
    print('re-encoded {} rows; jpeg columns: {}'.format(rows, cols))
    return 0


if __name__ == '__main__':
    sys.exit(main())
