"""Worker->main payload serializers for the process pool.

Parity: /root/reference/petastorm/reader_impl/pickle_serializer.py:17-23 and
arrow_table_serializer.py:22-33.
"""

import pickle


class PickleSerializer(object):
    """Row-dict payloads (reference pickle_serializer.py:17-23)."""

    def serialize(self, rows):
        return pickle.dumps(rows, protocol=pickle.HIGHEST_PROTOCOL)

    def deserialize(self, data):
        return pickle.loads(data)


class ArrowTableSerializer(object):
    """Arrow-table payloads via the Arrow IPC stream format
    (reference arrow_table_serializer.py:22-33)."""

    def serialize(self, table):
        import pyarrow as pa
        sink = pa.BufferOutputStream()
        with pa.ipc.new_stream(sink, table.schema) as writer:
            writer.write_table(table)
        return sink.getvalue().to_pybytes()

    def deserialize(self, data):
        import pyarrow as pa
        with pa.ipc.open_stream(data) as reader:
            return reader.read_all()
