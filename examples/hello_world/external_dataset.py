"""Read a NON-petastorm ("external") Parquet store with make_batch_reader.

Mirrors the reference's examples/hello_world/external_dataset flow: any
plain Parquet dataset — written by pyarrow, Spark, or anything else —
reads back as column batches with no petastorm metadata required.
With --device cuda the same store decodes through the MI355X HIP
pipeline (snappy pages, delta/dict encodings, strings included).
"""
import argparse
import os
import sys
import tempfile

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))

import numpy as np


def generate_external_dataset(path, num_rows=200):
    import pyarrow as pa
    import pyarrow.parquet as pq
    rng = np.random.RandomState(0)
    table = pa.table({
        'id': np.arange(num_rows, dtype=np.int64),
        'value1': rng.rand(num_rows),
        'value2': rng.randint(0, 100, num_rows).astype(np.int32),
        'name': np.array(['sensor_%d' % (i % 5) for i in range(num_rows)]),
    })
    pq.write_table(table, path + '/data.parquet', row_group_size=50)


def python_hello_world(url, device=None):
    from petastorm_amd import make_batch_reader
    kwargs = {'device': device} if device else {}
    with make_batch_reader(url, num_epochs=1, **kwargs) as reader:
        for batch in reader:
            ids = batch.id
            print('batch of', len(ids), 'rows; first id:',
                  int(ids[0]), 'name:', batch.name[0])


def pytorch_hello_world(url, device=None):
    from petastorm_amd import make_batch_reader
    from petastorm_amd.pytorch import BatchedDataLoader
    kwargs = {'device': device} if device else {}
    reader = make_batch_reader(url, num_epochs=1,
                               schema_fields=['id', 'value1'], **kwargs)
    with BatchedDataLoader(reader, batch_size=64) as loader:
        batch = next(iter(loader))
        print('pytorch batch:', batch['id'].shape, batch['id'][:5])


if __name__ == '__main__':
    parser = argparse.ArgumentParser(description=__doc__)
    parser.add_argument('url', nargs='?', default=None)
    parser.add_argument('--device', default=None,
                        help="e.g. 'cuda' for the MI355X decode pipeline")
    args = parser.parse_args()
    url = args.url
    if url is None:
        d = tempfile.mkdtemp(prefix='external_ds_')
        generate_external_dataset(d)
        url = 'file://' + d
    python_hello_world(url, args.device)
    pytorch_hello_world(url, args.device)
