"""Dump dataset schema / row-group layout / indexes.

Parity: /root/reference/petastorm/etl/metadata_util.py:16-70.
"""

import argparse
import sys

from petastorm_amd.etl.dataset_metadata import (infer_or_load_unischema,
                                                load_row_groups)
from petastorm_amd.etl.rowgroup_indexing import load_rowgroup_indexes
from petastorm_amd.fs_utils import get_filesystem_and_path_or_paths


def main(args=None):
    ap = argparse.ArgumentParser(description='Inspect a petastorm_amd '
                                             'dataset')
    ap.add_argument('dataset_url')
    ap.add_argument('--print-schema', action='store_true')
    ap.add_argument('--print-row-groups', action='store_true')
    ap.add_argument('--print-index', action='store_true')
    a = ap.parse_args(args)
    fs, path = get_filesystem_and_path_or_paths(a.dataset_url)

    if a.print_schema:
        schema, stored = infer_or_load_unischema(fs, path)
        print('schema ({}):'.format('stored' if stored else 'inferred'))
        print(schema)
    if a.print_row_groups:
        pieces = load_row_groups(fs, path)
        print('{} row groups:'.format(len(pieces)))
        for p in pieces:
            print('  [{:4d}] {}#{} rows={}'.format(p.index, p.path,
                                                   p.row_group, p.num_rows))
    if a.print_index:
        indexes = load_rowgroup_indexes(fs, path)
        if not indexes:
            print('no row-group indexes')
        for name, ix in indexes.items():
            print('index {!r} over {}: {} distinct values'
                  .format(name, ix.column_names, len(ix.indexed_values)))
    return 0


if __name__ == '__main__':
    sys.exit(main())
