"""Dataset write path + schema/row-group metadata.

Parity: /root/reference/petastorm/etl/dataset_metadata.py.

* ``materialize_dataset`` context manager (reference :52-132) — here it
  yields a :class:`DatasetWriter` (the reference relies on Spark executors
  inside the context; this framework writes through pyarrow directly).
  ``rowgroup_size_mb`` plays the role of ``parquet.block.size``
  (reference :177-178).
* Schema storage: a JSON descriptor written both (a) into every Parquet
  file's key-value metadata under ``UNISCHEMA_KEY`` and (b) as a
  ``_petastorm_amd_metadata.json`` sidecar for O(1) discovery — replacing
  the reference's *pickled* Unischema in ``_common_metadata``
  (reference :194-205; pickle fragility acknowledged there at :202-204).
* ``load_row_groups`` returns sorted per-row-group pieces (reference
  :244-290, sorted order for reproducibility at :276-278).
* ``get_schema`` / ``get_schema_from_dataset_url`` /
  ``infer_or_load_unischema`` (reference :356-418).
"""

import json
import os
import posixpath
from collections import namedtuple
from contextlib import contextmanager

from petastorm_amd.errors import (PetastormMetadataError,  # noqa: F401
                                  PetastormMetadataGenerationError)
from petastorm_amd.fs_utils import get_filesystem_and_path_or_paths
from petastorm_amd.unischema import Unischema, dict_to_encoded_row

UNISCHEMA_KEY = 'petastorm_amd.unischema.v1'
METADATA_FILENAME = '_petastorm_amd_metadata.json'
DEFAULT_ROWGROUP_SIZE_MB = 32

#: One Parquet row group: the framework's unit of IO, sharding and shuffling.
#: ``index`` is the global ordinal across the sorted file list (the value
#: sharding is computed over, reference reader.py:573-597).  ``partitions``
#: holds hive-style ``key=value`` directory values for partitioned stores
#: (reference supports these through pq.ParquetDataset partitions,
#: arrow_reader_worker.py:358).
RowGroupPiece = namedtuple('RowGroupPiece',
                           ['index', 'path', 'row_group', 'num_rows',
                            'partitions'])
RowGroupPiece.__new__.__defaults__ = ({},)


def parse_partition_values(root, file_path):
    """Extract hive-style key=value path segments between ``root`` and the
    file: .../root/color=red/size=2/part.parquet -> {'color': 'red',
    'size': '2'}."""
    rel = file_path[len(root):].lstrip('/')
    out = {}
    for seg in rel.split('/')[:-1]:
        if '=' in seg:
            k, v = seg.split('=', 1)
            out[k] = v
    return out


class DatasetWriter(object):
    """Encodes row dicts with the schema's codecs and writes Parquet files
    with bounded row-group byte size.

    A row group is flushed when its encoded payload reaches
    ``rowgroup_size_mb`` (the reference controls the same thing through
    Spark's ``parquet.block.size``, dataset_metadata.py:177-178).
    """

    def __init__(self, fs, path, schema, rowgroup_size_mb=DEFAULT_ROWGROUP_SIZE_MB,
                 compression='snappy', file_prefix='part',
                 rows_per_rowgroup=None):
        self._fs = fs
        self._path = path
        self._schema = schema
        self._rowgroup_bytes = int(rowgroup_size_mb * (1 << 20))
        self._rows_per_rowgroup = rows_per_rowgroup
        self._compression = compression
        self._file_prefix = file_prefix

        self._arrow_schema = schema.as_arrow_schema().with_metadata(
            {UNISCHEMA_KEY: schema.to_json()})
        self._buffer = []          # encoded row dicts
        self._buffer_bytes = 0
        self._writer = None
        self._file_index = 0
        self._rows_per_file = None
        self._rows_in_file = 0

    # ------------------------------------------------------------------
    def write_row(self, row_dict):
        encoded = dict_to_encoded_row(self._schema, row_dict)
        self._buffer.append(encoded)
        self._buffer_bytes += self._estimate_row_bytes(encoded)
        if self._rows_per_rowgroup is not None:
            # exact-rows mode: uniform row groups (equal shards for
            # distributed readers; see parallel/epochs.py ordering contract)
            if len(self._buffer) >= self._rows_per_rowgroup:
                self._flush_row_group()
        elif self._buffer_bytes >= self._rowgroup_bytes:
            self._flush_row_group()

    def write_rows(self, row_dicts):
        for r in row_dicts:
            self.write_row(r)

    def new_file(self):
        """Close the current file; subsequent rows go to a fresh file."""
        self._flush_row_group()
        if self._writer is not None:
            self._writer.close()
            self._writer = None

    @staticmethod
    def _estimate_row_bytes(encoded):
        total = 0
        for v in encoded.values():
            if isinstance(v, (bytes, bytearray)):
                total += len(v)
            elif isinstance(v, str):
                total += len(v)
            else:
                total += 8
        return total

    def _flush_row_group(self):
        if not self._buffer:
            return
        import pyarrow as pa
        cols = {}
        for f in self._arrow_schema.names:
            cols[f] = [row[f] for row in self._buffer]
        table = pa.Table.from_pydict(cols, schema=self._arrow_schema)
        if self._writer is None:
            import pyarrow.parquet as pq
            fname = '{}-{:05d}.parquet'.format(self._file_prefix, self._file_index)
            self._file_index += 1
            full = posixpath.join(self._path, fname)
            self._writer = pq.ParquetWriter(
                self._fs.open(full, 'wb'), self._arrow_schema,
                compression=self._column_compression(),
                # one table write == one row group:
                use_dictionary=False, write_statistics=True,
                # 256 KiB pages: the page is the GPU decode-parallelism unit
                # (one wave per page), so smaller-than-arrow-default pages
                # keep the 256-CU chip fed on few-column datasets
                # (sweepable via PSA_WRITER_PAGE_KB)
                data_page_size=int(os.environ.get(
                    'PSA_WRITER_PAGE_KB', '256')) << 10)
        self._writer.write_table(table)
        self._buffer = []
        self._buffer_bytes = 0

    def _column_compression(self):
        """Per-column codec choice: columns whose payloads are already
        compressed (jpeg/png images, zlib ndarrays) are stored UNCOMPRESSED —
        re-snappy-ing them wastes CPU at write time AND, for images, lets the
        MI355X jpeg decoder parse headers straight from the pinned host
        buffer without a decompress round-trip."""
        from petastorm_amd.codecs import (CompressedImageCodec,
                                          CompressedNdarrayCodec)
        if self._compression == 'none':
            return 'none'
        spec = {}
        for f in self._schema.fields.values():
            if isinstance(f.codec, (CompressedImageCodec,
                                    CompressedNdarrayCodec)):
                spec[f.name] = 'none'
            else:
                spec[f.name] = self._compression
        return spec

    def close(self):
        self._flush_row_group()
        if self._writer is not None:
            self._writer.close()
            self._writer = None


@contextmanager
def materialize_dataset(dataset_url, schema,
                        rowgroup_size_mb=DEFAULT_ROWGROUP_SIZE_MB,
                        compression='snappy', filesystem_factory=None,
                        rows_per_rowgroup=None):
    """Write a dataset: ``with materialize_dataset(url, schema) as writer: ...``

    On exit, schema metadata is persisted next to the data (reference
    materialize_dataset, dataset_metadata.py:52-132).
    """
    fs, path = get_filesystem_and_path_or_paths(dataset_url)
    fs.makedirs(path, exist_ok=True)
    writer = DatasetWriter(fs, path, schema, rowgroup_size_mb, compression,
                           rows_per_rowgroup=rows_per_rowgroup)
    try:
        yield writer
    except BaseException:
        # close file handles, but do NOT stamp schema metadata: a
        # half-written store must not look complete to get_schema
        writer.close()
        raise
    writer.close()
    _write_dataset_metadata(fs, path, schema)


def _write_dataset_metadata(fs, path, schema):
    meta = {
        'version': 1,
        'created_by': 'petastorm_amd',
        'unischema': json.loads(schema.to_json()),
    }
    with fs.open(posixpath.join(path, METADATA_FILENAME), 'w') as f:
        f.write(json.dumps(meta, indent=2))


# ---------------------------------------------------------------------------
# read side
# ---------------------------------------------------------------------------

def list_parquet_files(fs, path_or_paths):
    """All data files of the dataset, sorted (reproducible piece order,
    reference :276-278)."""
    paths = path_or_paths if isinstance(path_or_paths, list) else [path_or_paths]
    files = []
    for p in paths:
        if fs.isdir(p):
            for f in fs.find(p):
                base = posixpath.basename(f)
                if base.endswith('.parquet') and not base.startswith('_') \
                        and not base.startswith('.'):
                    files.append(f)
        else:
            files.append(p)
    return sorted(files)


def load_row_groups(fs, path_or_paths):
    """Enumerate every row group of the dataset as RowGroupPieces
    (reference load_row_groups, :244-290).  Hive-partition directory values
    are attached to each piece."""
    import pyarrow.parquet as pq
    roots = path_or_paths if isinstance(path_or_paths, list) \
        else [path_or_paths]
    dir_roots = [r for r in roots if fs.isdir(r)]
    pieces = []
    index = 0
    for fpath in list_parquet_files(fs, path_or_paths):
        partitions = {}
        for r in dir_roots:
            if fpath.startswith(r.rstrip('/') + '/'):
                partitions = parse_partition_values(r.rstrip('/'), fpath)
                break
        with fs.open(fpath, 'rb') as f:
            md = pq.ParquetFile(f).metadata
        for rg in range(md.num_row_groups):
            pieces.append(RowGroupPiece(index, fpath, rg,
                                        md.row_group(rg).num_rows,
                                        partitions))
            index += 1
    return pieces


def get_schema(fs, path_or_paths):
    """Load the stored Unischema. Raises ValueError when the dataset was not
    written by this framework (reference get_schema, :356-385)."""
    paths = path_or_paths if isinstance(path_or_paths, list) else [path_or_paths]
    # 1) sidecar
    for p in paths:
        candidate = posixpath.join(p, METADATA_FILENAME) if fs.isdir(p) \
            else posixpath.join(posixpath.dirname(p), METADATA_FILENAME)
        if fs.exists(candidate):
            with fs.open(candidate, 'r') as f:
                meta = json.loads(f.read())
            return Unischema.from_json(json.dumps(meta['unischema']))
    # 2) per-file key-value metadata
    import pyarrow.parquet as pq
    files = list_parquet_files(fs, path_or_paths)
    if files:
        with fs.open(files[0], 'rb') as f:
            md = pq.ParquetFile(f).schema_arrow.metadata
        if md and UNISCHEMA_KEY.encode() in md:
            return Unischema.from_json(md[UNISCHEMA_KEY.encode()].decode())
    # 3) upstream-petastorm store: pickled Unischema in _common_metadata
    # (read-only interop; reference etl/dataset_metadata.py:356-385)
    from petastorm_amd.etl import interop
    schema = interop.load_reference_unischema(fs, paths)
    if schema is not None:
        return schema
    raise ValueError('Dataset at {} has no petastorm_amd schema metadata. Use '
                     'make_batch_reader for plain Parquet stores.'
                     .format(path_or_paths))


def get_schema_from_dataset_url(dataset_url_or_urls, storage_options=None):
    fs, path_or_paths = get_filesystem_and_path_or_paths(
        dataset_url_or_urls, storage_options)
    return get_schema(fs, path_or_paths)


def infer_or_load_unischema(fs, path_or_paths):
    """Stored schema when present, else inference from the Arrow schema
    (reference infer_or_load_unischema, :410-418)."""
    import numpy as _np

    from petastorm_amd.unischema import UnischemaField

    def _with_partition_fields(schema):
        pieces = load_row_groups(fs, path_or_paths)
        part_keys = sorted({k for p in pieces for k in p.partitions})
        extra = []
        for k in part_keys:
            if k in schema.fields:
                continue
            values = {p.partitions.get(k) for p in pieces}
            try:
                all(int(v) for v in values if v is not None)
                dtype = _np.int64
            except (TypeError, ValueError):
                dtype = _np.str_
            extra.append(UnischemaField(k, dtype, (), None, False))
        if not extra:
            return schema
        return Unischema(schema._name,
                         list(schema.fields.values()) + extra)

    try:
        return _with_partition_fields(get_schema(fs, path_or_paths)), True
    except ValueError:
        import pyarrow.parquet as pq
        files = list_parquet_files(fs, path_or_paths)
        if not files:
            raise ValueError('No parquet files found at {}'.format(path_or_paths))
        with fs.open(files[0], 'rb') as f:
            arrow_schema = pq.ParquetFile(f).schema_arrow
        return _with_partition_fields(
            Unischema.from_arrow_schema(arrow_schema)), False


def select_pieces_by_filters(fs, pieces, filters):
    """Statistics/partition-based row-group pruning for pyarrow-style
    ``filters`` (the reference forwards these to pq.ParquetDataset,
    reference reader.py:431-433 ``filters`` kwarg).

    ``filters`` is a list of ``(column, op, value)`` tuples (ANDed), or a
    list of such lists (ORed DNF).  Ops: ==, =, !=, <, <=, >, >=, in,
    not in.  A row group is kept unless its column-chunk min/max statistics
    (or hive partition value) PROVE no row can match — unknown statistics
    keep the group (pruning is an optimization, never a correctness filter).
    """
    import pyarrow.parquet as pq
    if not filters:
        return pieces
    if filters and isinstance(filters[0], tuple):
        dnf = [filters]
    else:
        dnf = filters

    stats_cache = {}

    def col_range(piece, name):
        if name in piece.partitions:
            v = piece.partitions[name]
            return v, v
        key = piece.path
        if key not in stats_cache:
            with fs.open(key, 'rb') as f:
                stats_cache[key] = pq.ParquetFile(f).metadata
        md = stats_cache[key].row_group(piece.row_group)
        for ci in range(md.num_columns):
            col = md.column(ci)
            if col.path_in_schema == name:
                st = col.statistics
                if st is not None and st.has_min_max:
                    return st.min, st.max
        return None, None

    def clause_may_match(piece, clause):
        for name, op, value in clause:
            lo, hi = col_range(piece, name)
            if lo is None:
                continue  # unknown -> can't prune on this term
            try:
                if op in ('==', '='):
                    if not (lo <= value <= hi):
                        return False
                elif op == '<':
                    if not (lo < value):
                        return False
                elif op == '<=':
                    if not (lo <= value):
                        return False
                elif op == '>':
                    if not (hi > value):
                        return False
                elif op == '>=':
                    if not (hi >= value):
                        return False
                elif op == 'in':
                    if not any(lo <= v <= hi for v in value):
                        return False
                elif op in ('!=', 'not in'):
                    # can only prune when the whole range is one value
                    excluded = [value] if op == '!=' else list(value)
                    if lo == hi and lo in excluded:
                        return False
            except TypeError:
                continue  # incomparable types: keep
        return True

    return [p for p in pieces
            if any(clause_may_match(p, clause) for clause in dnf)]
