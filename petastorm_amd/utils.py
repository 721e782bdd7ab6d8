"""Small shared helpers.

Parity: /root/reference/petastorm/utils.py (decode_row :52-85).  The
metadata read-modify-write helper lives in petastorm_amd/etl/dataset_metadata.py
since our metadata is JSON sidecar files, not pickled parquet key-values.
"""

from petastorm_amd import codecs as _codecs


def decode_row(row, schema):
    """Decode all fields of a stored row dict using the schema's codecs.

    Reference: petastorm/utils.py:52-85.  ``None`` values stay ``None``
    (nullable fields).
    """
    decoded = {}
    for name, value in row.items():
        field = schema.fields.get(name)
        if field is None:
            # columns not in the schema view pass through untouched
            decoded[name] = value
            continue
        decoded[name] = _codecs.decode_value(field, value)
    return decoded


def run_in_subprocess(func, *args, **kwargs):
    """Run ``func(*args, **kwargs)`` in a spawned subprocess and return its
    result (reference petastorm/utils.py:28-45).

    Used to isolate work that must not pollute the parent (e.g. library
    global state).  The callable and its arguments must be picklable.
    """
    import multiprocessing as mp
    ctx = mp.get_context('spawn')
    with ctx.Pool(1) as pool:
        return pool.apply(func, args, kwargs)


#: Re-exported for drop-in compatibility with ``petastorm.utils``
#: (reference utils.py:48-49).
from petastorm_amd.errors import DecodeFieldError  # noqa: E402,F401


def add_to_dataset_metadata(dataset_url_or_path, key, value):
    """Read-modify-write one key of the dataset's ``_common_metadata``
    key-value metadata (reference petastorm/utils.py:88-132 — there it
    carries the pickled schema; here JSON sidecars carry OURS, but
    upstream-interop stores still use _common_metadata, so this helper
    operates on that file for them)."""
    import posixpath

    import pyarrow as pa
    import pyarrow.parquet as pq

    from petastorm_amd.etl.dataset_metadata import list_parquet_files
    from petastorm_amd.fs_utils import get_filesystem_and_path_or_paths
    fs, path = get_filesystem_and_path_or_paths(dataset_url_or_path)
    common = posixpath.join(path, '_common_metadata')
    if fs.exists(common):
        with fs.open(common, 'rb') as f:
            schema = pq.read_schema(f)
    else:
        files = list_parquet_files(fs, path)
        if not files:
            raise ValueError('No parquet files at {}'.format(path))
        with fs.open(files[0], 'rb') as f:
            schema = pq.ParquetFile(f).schema_arrow
    md = dict(schema.metadata or {})
    md[key if isinstance(key, bytes) else key.encode()] = \
        value if isinstance(value, bytes) else str(value).encode()
    with fs.open(common, 'wb') as f:
        pq.write_metadata(schema.with_metadata(md), f)
