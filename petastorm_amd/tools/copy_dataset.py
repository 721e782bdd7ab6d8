"""Copy a petastorm_amd dataset, optionally subsetting columns, filtering
nulls and re-chunking row groups.

Parity: /root/reference/petastorm/tools/copy_dataset.py:34-91 (which runs the
copy on Spark executors; this framework streams row-groups through the
reader/writer pair — no Spark in the serving environment).
"""

import argparse
import sys

from petastorm_amd import make_reader
from petastorm_amd.etl.dataset_metadata import materialize_dataset
from petastorm_amd.unischema import Unischema


def copy_dataset(source_url, target_url, field_regex=None,
                 not_null_fields=None, rowgroup_size_mb=32,
                 compression='snappy'):
    """Stream-copy ``source_url`` to ``target_url``.

    :param field_regex: optional list of regex patterns selecting columns
    :param not_null_fields: drop rows where any of these fields is null
    """
    with make_reader(source_url, schema_fields=field_regex,
                     shuffle_row_groups=False, reader_pool_type='thread',
                     workers_count=4) as reader:
        out_schema = Unischema(
            'copied', list(reader.schema.fields.values()))
        with materialize_dataset(target_url, out_schema, rowgroup_size_mb,
                                 compression) as writer:
            copied = 0
            for row in reader:
                d = row._asdict()
                if not_null_fields and any(d.get(f) is None
                                           for f in not_null_fields):
                    continue
                writer.write_row(d)
                copied += 1
    return copied


def main(args=None):
    ap = argparse.ArgumentParser(description='Copy a petastorm_amd dataset')
    ap.add_argument('source_url')
    ap.add_argument('target_url')
    ap.add_argument('--field-regex', nargs='+', default=None)
    ap.add_argument('--not-null-fields', nargs='+', default=None)
    ap.add_argument('--rowgroup-size-mb', type=int, default=32)
    ap.add_argument('--compression', default='snappy')
    a = ap.parse_args(args)
    n = copy_dataset(a.source_url, a.target_url, a.field_regex,
                     a.not_null_fields, a.rowgroup_size_mb, a.compression)
    print('copied {} rows'.format(n))
    return 0


if __name__ == '__main__':
    sys.exit(main())
