"""Add petastorm_amd schema metadata to an existing Parquet store.

Parity: /root/reference/petastorm/etl/petastorm_generate_metadata.py:47-111
(``petastorm-generate-metadata.py``).  Useful when a dataset was produced by
another writer (plain pyarrow, Spark, ...) and should afterwards be readable
through ``make_reader`` with full Unischema semantics.

The schema source is either (a) a user-supplied Unischema (import path), or
(b) inference from the store's Arrow schema (scalars/lists only).
"""

import argparse
import importlib
import sys

from petastorm_amd.etl.dataset_metadata import (_write_dataset_metadata,
                                                infer_or_load_unischema)
from petastorm_amd.fs_utils import get_filesystem_and_path_or_paths


def generate_metadata(dataset_url, unischema_class=None):
    """:param unischema_class: 'module.path.SchemaObject' or None to infer"""
    fs, path = get_filesystem_and_path_or_paths(dataset_url)
    if unischema_class:
        module_path, name = unischema_class.rsplit('.', 1)
        schema = getattr(importlib.import_module(module_path), name)
    else:
        schema, stored = infer_or_load_unischema(fs, path)
    _write_dataset_metadata(fs, path, schema)
    return schema


def main(args=None):
    ap = argparse.ArgumentParser(
        description='Write petastorm_amd schema metadata next to an '
                    'existing Parquet store')
    ap.add_argument('dataset_url')
    ap.add_argument('--unischema-class', default=None,
                    help="e.g. 'mypackage.schemas.MySchema'")
    a = ap.parse_args(args)
    schema = generate_metadata(a.dataset_url, a.unischema_class)
    print('wrote metadata for schema with fields: {}'
          .format(sorted(schema.fields)))
    return 0


if __name__ == '__main__':
    sys.exit(main())


#: Reference-name alias (petastorm_generate_metadata.py:47).
generate_petastorm_metadata = generate_metadata
