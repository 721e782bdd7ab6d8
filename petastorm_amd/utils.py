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
