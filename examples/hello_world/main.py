"""HelloWorld: generate a small petastorm_amd dataset and read it back.

Parity role: /root/reference/examples/hello_world/petastorm_dataset/
(generate_petastorm_dataset.py + python_hello_world.py + pytorch_hello_world.py).

Run:  python examples/hello_world/main.py [output_url]
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))

import tempfile

import numpy as np

from petastorm_amd import make_reader
from petastorm_amd.codecs import (CompressedImageCodec, NdarrayCodec,
                                  ScalarCodec)
from petastorm_amd.etl.dataset_metadata import materialize_dataset
from petastorm_amd.pytorch import DataLoader
from petastorm_amd.unischema import Unischema, UnischemaField

HelloWorldSchema = Unischema('HelloWorldSchema', [
    UnischemaField('id', np.int32, (), ScalarCodec(), False),
    UnischemaField('image1', np.uint8, (128, 256, 3),
                   CompressedImageCodec('png'), False),
    UnischemaField('array_4d', np.uint8, (None, 128, 30, None),
                   NdarrayCodec(), False),
])


def row_generator(i):
    return {
        'id': np.int32(i),
        'image1': np.random.randint(0, 255, (128, 256, 3), dtype=np.uint8),
        'array_4d': np.random.randint(0, 255, (4, 128, 30, 3),
                                      dtype=np.uint8),
    }


def generate_dataset(url, rows=30):
    with materialize_dataset(url, HelloWorldSchema, rowgroup_size_mb=8) as w:
        for i in range(rows):
            w.write_row(row_generator(i))
    print('wrote {} rows to {}'.format(rows, url))


def python_hello_world(url):
    with make_reader(url) as reader:
        for row in reader:
            print('id={} image1.shape={} array_4d.shape={}'.format(
                row.id, row.image1.shape, row.array_4d.shape))
            break


def pytorch_hello_world(url):
    reader = make_reader(url, schema_fields=['id'], shuffle_row_groups=False)
    with DataLoader(reader, batch_size=10) as loader:
        batch = next(iter(loader))
        print('pytorch batch id tensor:', batch['id'][:5])


if __name__ == '__main__':
    import argparse
    parser = argparse.ArgumentParser(description=__doc__)
    parser.add_argument('url', nargs='?',
                        default='file://' + tempfile.mkdtemp(prefix='hello_world_'),
                        help='dataset output URL (default: fresh temp dir)')
    url = parser.parse_args().url
    generate_dataset(url)
    python_hello_world(url)
    pytorch_hello_world(url)
