"""Row-dict decode worker: the ``make_reader`` decode stage.

Parity: /root/reference/petastorm/py_dict_reader_worker.py.

Per ventilated row-group the worker:

1. consults the read-through cache (reference :164-169)
2. reads the row group through pyarrow (the CPU path deliberately delegates
   Parquet page decode to Arrow C++, exactly like the reference :264-268;
   the MI355X-native page decode lives on the GPU batch path)
3. predicate-first two-phase column load (reference :197-262)
4. per-row codec decode via ``decode_row`` (reference :190)
5. optional in-row-group shuffle + shuffle-row-drop partitioning
   (reference :264-286)
6. per-row TransformSpec (reference :38-52)
7. NGram window assembly (reference :171-172)
8. publishes the decoded row list
"""

import hashlib

import numpy as np

from petastorm_amd.utils import decode_row
from petastorm_amd.workers_pool.worker_base import WorkerBase


class RowWorkerArgs(object):
    """Picklable bundle of worker construction args."""

    def __init__(self, fs, schema, view_schema, ngram, pieces, cache,
                 transform_spec, transformed_schema, shuffle_rows=False,
                 seed=None):
        self.fs = fs
        self.schema = schema                  # full storage schema
        self.view_schema = view_schema        # selected-fields view
        self.ngram = ngram
        self.pieces = pieces                  # list of RowGroupPiece
        self.cache = cache
        self.transform_spec = transform_spec
        self.transformed_schema = transformed_schema
        self.shuffle_rows = shuffle_rows
        self.seed = seed


def _partition_value(field, raw):
    """Cast a hive path value to the field's dtype."""
    if field is None or field.numpy_dtype in (np.str_,):
        return str(raw)
    return np.dtype(field.numpy_dtype).type(raw)


def _cache_key(path, row_group, column_names):
    h = hashlib.md5('{}:{}:{}'.format(path, row_group,
                                      ','.join(sorted(column_names)))
                    .encode('utf-8')).hexdigest()
    return 'rg-{}'.format(h)


class RowReaderWorker(WorkerBase):
    def __init__(self, worker_id, publish_func, args):
        super(RowReaderWorker, self).__init__(worker_id, publish_func, args)
        self._a = args
        self._parquet_files = {}

    # ------------------------------------------------------------------
    def process(self, piece_index, worker_predicate=None,
                shuffle_row_drop_partition=(0, 1)):
        piece = self._a.pieces[piece_index]
        if worker_predicate is not None:
            rows = self._load_rows_with_predicate(piece, worker_predicate)
        else:
            needed = list(self._a.view_schema.fields.keys())
            key = _cache_key(piece.path, piece.row_group, needed)
            rows = self._a.cache.get(
                key, lambda: self._load_rows(piece, needed))
        rows = self._shuffle_and_drop(rows, piece,
                                      shuffle_row_drop_partition)
        rows = [decode_row(r, self._a.view_schema) for r in rows]
        if self._a.transform_spec is not None and self._a.transform_spec.func:
            # a func may return None to drop the row (the row-path analog
            # of the batch path's row-filtering transforms)
            rows = [t for t in (self._a.transform_spec.func(r)
                                for r in rows) if t is not None]
        if self._a.transform_spec is not None:
            keep = set(self._a.transformed_schema.fields.keys())
            rows = [{k: v for k, v in r.items() if k in keep} for r in rows]
        if self._a.ngram is not None:
            rows = self._a.ngram.form_ngram(rows, self._a.transformed_schema)
        if rows:
            self.publish_func(rows)

    # ------------------------------------------------------------------
    def shutdown(self):
        # close cached footer/file handles (held per worker for the
        # pool's lifetime); without this they linger until GC
        for pf in self._parquet_files.values():
            try:
                pf.close()
            except Exception:  # noqa: BLE001 - best-effort teardown
                pass
        self._parquet_files.clear()

    def _parquet_file(self, path):
        if path not in self._parquet_files:
            import pyarrow.parquet as pq
            self._parquet_files[path] = pq.ParquetFile(
                self._a.fs.open(path, 'rb'))
        return self._parquet_files[path]

    def _load_rows(self, piece, column_names):
        """Read one row group as a list of raw (encoded) row dicts.
        Hive-partition key columns are materialized from the piece's path
        values (reference reads them through pq.ParquetDataset partitions,
        py_dict_reader_worker.py:267)."""
        pf = self._parquet_file(piece.path)
        available = set(pf.schema_arrow.names)
        cols = [c for c in column_names if c in available]
        table = pf.read_row_group(piece.row_group, columns=cols)
        pydict = table.to_pydict()
        names = list(pydict.keys())
        rows = [dict(zip(names, vals)) for vals in zip(*pydict.values())] \
            if names else [{} for _ in range(table.num_rows)]
        part_cols = [c for c in column_names
                     if c not in available and c in piece.partitions]
        for c in part_cols:
            v = _partition_value(self._a.schema.fields.get(c),
                                 piece.partitions[c])
            for r in rows:
                r[c] = v
        return rows

    def _load_rows_with_predicate(self, piece, predicate):
        """Two-phase load: predicate columns first, then the rest only for
        matching rows (reference py_dict_reader_worker.py:197-262)."""
        predicate_fields = list(predicate.get_fields())
        all_fields = list(self._a.view_schema.fields.keys())
        other_fields = [f for f in all_fields if f not in predicate_fields]

        pf = self._parquet_file(piece.path)
        available = set(pf.schema_arrow.names)
        part_pred = [f for f in predicate_fields
                     if f not in available and f in piece.partitions]
        predicate_fields = [f for f in predicate_fields if f in available]
        part_vals = {f: _partition_value(self._a.schema.fields.get(f),
                                         piece.partitions[f])
                     for f in part_pred}
        if part_pred and not predicate_fields:
            # all predicate fields are partition keys: evaluate once
            if not predicate.do_include(part_vals):
                return []
            return self._load_rows(piece, all_fields)
        pred_table = pf.read_row_group(piece.row_group,
                                       columns=predicate_fields)
        pred_cols = {name: pred_table.column(name).to_pylist()
                     for name in predicate_fields}
        n = pred_table.num_rows
        # predicate operates on DECODED values (reference decodes predicate
        # columns before evaluating, :232); mixed predicates see their
        # partition-key fields too (constant within the row group)
        decoded_pred_rows = [
            dict(decode_row({f: pred_cols[f][i]
                             for f in predicate_fields}, self._a.schema),
                 **part_vals)
            for i in range(n)]
        match_idx = [i for i, r in enumerate(decoded_pred_rows)
                     if predicate.do_include(r)]
        if not match_idx:
            return []
        other_file = [f for f in other_fields if f in available]
        other_part = [f for f in other_fields
                      if f not in available and f in piece.partitions]
        rest_cols = {}
        if other_file:
            rest = pf.read_row_group(piece.row_group, columns=other_file)
            rest = rest.take(match_idx)
            rest_cols = {name: rest.column(name).to_pylist()
                         for name in other_file}
        other_part_vals = {
            f: _partition_value(self._a.schema.fields.get(f),
                                piece.partitions[f]) for f in other_part}
        rows = []
        for out_i, i in enumerate(match_idx):
            row = {f: pred_cols[f][i] for f in predicate_fields}
            row.update({f: rest_cols[f][out_i] for f in other_file})
            row.update(part_vals)
            row.update(other_part_vals)
            rows.append(row)
        return rows

    def _shuffle_and_drop(self, rows, piece, shuffle_row_drop_partition):
        part, num_parts = shuffle_row_drop_partition
        # rows are shuffled only when shuffle_rows is set; drop-partitioning
        # slices the stable order so the N partitions exactly cover the
        # row-group (reference py_dict_reader_worker.py:264-286)
        if self._a.shuffle_rows:
            rng = np.random.RandomState(
                None if self._a.seed is None
                else (self._a.seed + piece.index) % (2 ** 31))
            perm = rng.permutation(len(rows))
            rows = [rows[i] for i in perm]
        if num_parts > 1:
            # ngram windows need `length-1` rows of lookahead across the
            # partition boundary (reference :278-283)
            extension = (self._a.ngram.length - 1) if self._a.ngram else 0
            bounds = np.linspace(0, len(rows), num_parts + 1).astype(int)
            lo, hi = bounds[part], min(len(rows), bounds[part + 1] + extension)
            rows = rows[lo:hi]
        return rows
