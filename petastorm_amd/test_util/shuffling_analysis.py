"""Shuffle-quality measurement.

Parity: /root/reference/petastorm/test_util/shuffling_analysis.py:52-85 —
the correlation of the shuffled row-id sequence against natural order,
repeated over several runs, as a distribution.  A good shuffle has
correlations concentrated near 0.
"""

import numpy as np


def compute_correlation_distribution(dataset_url, id_column, shuffle_options,
                                     num_corr_samples=10):
    """Read ``dataset_url`` ``num_corr_samples`` times with the given
    shuffle options; return the abs Pearson correlation of the read order of
    ``id_column`` against sorted order for each run."""
    from petastorm_amd import make_reader

    correlations = []
    for _ in range(num_corr_samples):
        with make_reader(dataset_url, reader_pool_type='thread',
                         **shuffle_options) as reader:
            ids = np.asarray([getattr(row, id_column) for row in reader],
                             dtype=np.float64)
        natural = np.arange(len(ids), dtype=np.float64)
        if len(ids) < 2 or ids.std() == 0:
            correlations.append(0.0)
            continue
        corr = np.corrcoef(ids, natural)[0, 1]
        correlations.append(abs(float(corr)))
    return np.asarray(correlations)
