"""Schema-driven fake Reader for testing consumers without IO.

Parity: /root/reference/petastorm/test_util/reader_mock.py:19-82.
"""

import numpy as np


def schema_data_generator_example(schema):
    """Default value generator for a schema (reference reader_mock.py:68-82)."""
    row = {}
    rng = np.random.RandomState(0)
    for name, field in schema.fields.items():
        dt = np.dtype(field.numpy_dtype) if not isinstance(
            field.numpy_dtype, type(None)) else np.dtype(np.float64)
        if field.shape == () or field.shape is None:
            if dt.kind in 'iu':
                row[name] = dt.type(rng.randint(0, 100))
            elif dt.kind == 'f':
                row[name] = dt.type(rng.rand())
            elif dt.kind in 'SU':
                row[name] = 'mock'
            else:
                row[name] = dt.type(0)
        else:
            shape = tuple(d if d is not None else 3 for d in field.shape)
            if dt.kind in 'iu':
                row[name] = rng.randint(0, 100, shape).astype(dt)
            else:
                row[name] = rng.rand(*shape).astype(dt)
    return row


class ReaderMock(object):
    """A Reader-compatible object yielding generated rows forever
    (reference reader_mock.py:19-65)."""

    def __init__(self, schema, data_generator=schema_data_generator_example):
        self.schema = schema
        self.ngram = None
        self.batched_output = False
        self.last_row_consumed = False
        self._generator = data_generator

    def __iter__(self):
        return self

    def __next__(self):
        return self.schema.make_namedtuple(**self._generator(self.schema))

    next = __next__

    def reset(self):
        pass

    def stop(self):
        pass

    def join(self):
        pass

    @property
    def diagnostics(self):
        return {}
