"""ETL: dataset writing, metadata and row-group indexing.

Parity targets: /root/reference/petastorm/etl/ (dataset_metadata.py,
rowgroup_indexing.py, rowgroup_indexers.py).  The reference's write path runs
on PySpark executors; this framework writes through pyarrow directly (there
is no Spark in the MI355X serving environment) with the same observable
artifacts: a Parquet dataset whose row-group layout is controlled by
``rowgroup_size_mb`` plus schema metadata stored next to the data.
"""


class RowGroupIndexerBase(object):
    """Index builder/lookup protocol (reference etl/__init__.py:21-49)."""

    @property
    def index_name(self):
        raise NotImplementedError()

    @property
    def column_names(self):
        raise NotImplementedError()

    @property
    def indexed_values(self):
        raise NotImplementedError()

    def get_row_group_indexes(self, value_key):
        raise NotImplementedError()

    def build_index(self, decoded_rows, piece_index):
        raise NotImplementedError()
