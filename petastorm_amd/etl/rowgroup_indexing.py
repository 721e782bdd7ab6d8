"""Inverted row-group indexes: value -> {row-group ordinals}.

Parity: /root/reference/petastorm/etl/rowgroup_indexing.py (:37-81 build,
:136-158 load) and rowgroup_indexers.py (:21-75 SingleFieldIndexer, :78-124
FieldNotNullIndexer).

Differences by design: the reference builds indexes on Spark and stores them
*pickled* inside parquet ``_common_metadata``.  This framework builds them
in-process by scanning the dataset through the same worker machinery used
for reading, and stores them as a JSON sidecar
(``_petastorm_amd_indexes.json``) — no pickle, no Spark.  Index values are
JSON-encoded; practical key types are int/str/bool (the reference's md5/pickle
approach had the same practical envelope).
"""

import json
import posixpath

from petastorm_amd.etl import RowGroupIndexerBase
from petastorm_amd.etl import dataset_metadata as dsm
from petastorm_amd.fs_utils import get_filesystem_and_path_or_paths
from petastorm_amd.utils import decode_row

INDEXES_FILENAME = '_petastorm_amd_indexes.json'


class SingleFieldIndexer(RowGroupIndexerBase):
    """Index every distinct value of one field
    (reference rowgroup_indexers.py:21-75)."""

    def __init__(self, index_name, index_field):
        self._index_name = index_name
        self._column_name = index_field if isinstance(index_field, str) \
            else index_field.name
        self._index_data = {}

    @property
    def index_name(self):
        return self._index_name

    @property
    def column_names(self):
        return [self._column_name]

    @property
    def indexed_values(self):
        return list(self._index_data.keys())

    def get_row_group_indexes(self, value_key):
        return self._index_data.get(_key_str(value_key), set())

    def build_index(self, decoded_rows, piece_index):
        for row in decoded_rows:
            v = row[self._column_name]
            if v is None:
                continue
            self._index_data.setdefault(_key_str(v), set()).add(piece_index)

    # serialization
    def to_dict(self):
        return {'type': 'SingleFieldIndexer', 'index_name': self._index_name,
                'column': self._column_name,
                'data': {k: sorted(v) for k, v in self._index_data.items()}}

    @classmethod
    def from_dict(cls, d):
        obj = cls(d['index_name'], d['column'])
        obj._index_data = {k: set(v) for k, v in d['data'].items()}
        return obj


class FieldNotNullIndexer(RowGroupIndexerBase):
    """Index row groups that contain at least one non-null value of a field
    (reference rowgroup_indexers.py:78-124)."""

    _KEY = 'not_null'

    def __init__(self, index_name, index_field):
        self._index_name = index_name
        self._column_name = index_field if isinstance(index_field, str) \
            else index_field.name
        self._index_data = {self._KEY: set()}

    @property
    def index_name(self):
        return self._index_name

    @property
    def column_names(self):
        return [self._column_name]

    @property
    def indexed_values(self):
        return [self._KEY]

    def get_row_group_indexes(self, value_key=None):
        return self._index_data[self._KEY]

    def build_index(self, decoded_rows, piece_index):
        for row in decoded_rows:
            if row[self._column_name] is not None:
                self._index_data[self._KEY].add(piece_index)
                return

    def to_dict(self):
        return {'type': 'FieldNotNullIndexer', 'index_name': self._index_name,
                'column': self._column_name,
                'data': {k: sorted(v) for k, v in self._index_data.items()}}

    @classmethod
    def from_dict(cls, d):
        obj = cls(d['index_name'], d['column'])
        obj._index_data = {k: set(v) for k, v in d['data'].items()}
        return obj


_INDEXER_TYPES = {
    'SingleFieldIndexer': SingleFieldIndexer,
    'FieldNotNullIndexer': FieldNotNullIndexer,
}


def _key_str(value):
    """Canonical JSON-safe key for an indexed value."""
    if isinstance(value, bool):
        return 'b:{}'.format(value)
    if isinstance(value, (int,)):
        return 'i:{}'.format(value)
    if isinstance(value, float):
        return 'f:{}'.format(value)
    if isinstance(value, bytes):
        return 's:{}'.format(value.decode('utf-8', 'replace'))
    try:
        import numpy as _np
        if isinstance(value, _np.integer):
            return 'i:{}'.format(int(value))
        if isinstance(value, _np.floating):
            return 'f:{}'.format(float(value))
    except ImportError:  # pragma: no cover
        pass
    return 's:{}'.format(value)


def build_rowgroup_index(dataset_url, indexers):
    """Scan the dataset and persist the inverted indexes
    (reference rowgroup_indexing.py:37-81)."""
    fs, path = get_filesystem_and_path_or_paths(dataset_url)
    schema = dsm.get_schema(fs, path)
    pieces = dsm.load_row_groups(fs, path)
    import pyarrow.parquet as pq
    needed = sorted({c for ix in indexers for c in ix.column_names})
    for piece in pieces:
        with fs.open(piece.path, 'rb') as f:
            table = pq.ParquetFile(f).read_row_group(piece.row_group,
                                                     columns=needed)
        pydict = table.to_pydict()
        rows = [dict(zip(pydict.keys(), vals))
                for vals in zip(*pydict.values())]
        decoded = [decode_row(r, schema) for r in rows]
        for ix in indexers:
            ix.build_index(decoded, piece.index)
    payload = {'version': 1,
               'indexes': {ix.index_name: ix.to_dict() for ix in indexers}}
    base = path if fs.isdir(path) else posixpath.dirname(path)
    with fs.open(posixpath.join(base, INDEXES_FILENAME), 'w') as f:
        f.write(json.dumps(payload))
    return indexers


def load_rowgroup_indexes(fs, path_or_paths):
    """Load all stored indexes as {name: indexer}
    (reference rowgroup_indexing.py:136-158)."""
    paths = path_or_paths if isinstance(path_or_paths, list) else [path_or_paths]
    for p in paths:
        base = p if fs.isdir(p) else posixpath.dirname(p)
        candidate = posixpath.join(base, INDEXES_FILENAME)
        if fs.exists(candidate):
            with fs.open(candidate, 'r') as f:
                payload = json.loads(f.read())
            out = {}
            for name, d in payload['indexes'].items():
                out[name] = _INDEXER_TYPES[d['type']].from_dict(d)
            return out
    return {}


def get_row_group_indexes(dataset_url_or_fs, path_or_paths=None):
    """Load the dataset's persisted rowgroup indexes (reference
    rowgroup_indexing.py:136-158).  Accepts a URL, or (fs, path)."""
    if path_or_paths is None:
        from petastorm_amd.fs_utils import get_filesystem_and_path_or_paths
        fs, path_or_paths = get_filesystem_and_path_or_paths(
            dataset_url_or_fs)
    else:
        fs = dataset_url_or_fs
    return load_rowgroup_indexes(fs, path_or_paths)
