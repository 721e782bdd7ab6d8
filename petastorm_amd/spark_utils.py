"""Read a petastorm_amd dataset as a Spark RDD of decoded namedtuples.

Parity: /root/reference/petastorm/spark_utils.py:23-52 (``dataset_as_rdd``).
pyspark is imported lazily.
"""


def dataset_as_rdd(dataset_url, spark_session, schema_fields=None):
    """:return: pyspark RDD over decoded rows (schema namedtuples)."""
    from petastorm_amd.etl.dataset_metadata import (get_schema, load_row_groups)
    from petastorm_amd.fs_utils import get_filesystem_and_path_or_paths

    fs, path = get_filesystem_and_path_or_paths(dataset_url)
    schema = get_schema(fs, path)
    pieces = load_row_groups(fs, path)
    view = schema.create_schema_view(schema_fields) if schema_fields \
        else schema
    field_names = list(view.fields.keys())
    sc = spark_session.sparkContext

    def read_piece(piece_index):
        # executor-side: open the row group and decode rows
        import pyarrow.parquet as pq

        from petastorm_amd.utils import decode_row
        piece = pieces[piece_index]
        pf = pq.ParquetFile(piece.path)
        table = pf.read_row_group(piece.row_group, columns=field_names)
        pydict = table.to_pydict()
        rows = [dict(zip(pydict.keys(), vals))
                for vals in zip(*pydict.values())]
        return [view.make_namedtuple(**decode_row(r, view)) for r in rows]

    return sc.parallelize(range(len(pieces)), len(pieces)) \
        .flatMap(read_piece)
