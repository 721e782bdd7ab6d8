"""Mix several readers by sampling probability.

Parity: /root/reference/petastorm/weighted_sampling_reader.py:26-115.
"""

import numpy as np


class WeightedSamplingReader(object):
    """Yields rows from N readers, picking the source reader per ``next()``
    by normalized probability (reference :26-81; cumulative distribution
    at :62)."""

    def __init__(self, readers, probabilities, seed=None):
        if len(readers) != len(probabilities):
            raise ValueError('readers and probabilities must have the same '
                             'length')
        if not readers:
            raise ValueError('At least one reader is required')
        self._readers = list(readers)
        p = np.asarray(probabilities, dtype=np.float64)
        if (p < 0).any() or p.sum() <= 0:
            raise ValueError('probabilities must be non-negative and not all '
                             'zero')
        self._cum = np.cumsum(p / p.sum())
        self._rng = np.random.RandomState(seed)

        first = self._readers[0]
        self.batched_output = first.batched_output
        self.ngram = first.ngram
        self.schema = first.schema
        for r in self._readers[1:]:
            if r.batched_output != self.batched_output:
                raise ValueError('All readers must have the same '
                                 'batched_output mode')
            if (r.ngram is None) != (self.ngram is None):
                raise ValueError('All readers must agree on ngram usage')
            if set(r.schema.fields.keys()) != set(self.schema.fields.keys()):
                raise ValueError('All readers must share the same schema '
                                 'field set')

    def __iter__(self):
        return self

    def __next__(self):
        # clamp: float rounding can leave cum[-1] a hair under 1.0, and a
        # draw above it would index past the last reader
        idx = min(int(np.searchsorted(self._cum, self._rng.uniform())),
                  len(self._readers) - 1)
        return next(self._readers[idx])

    next = __next__

    def stop(self):
        for r in self._readers:
            r.stop()

    def join(self):
        for r in self._readers:
            r.join()

    @property
    def last_row_consumed(self):
        return all(r.last_row_consumed for r in self._readers)

    def __enter__(self):
        return self

    def __exit__(self, exc_type, exc_val, exc_tb):
        self.stop()
        self.join()
