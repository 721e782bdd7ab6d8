"""Shuffle-quality measurement.

Parity: /root/reference/petastorm/test_util/shuffling_analysis.py —
both halves: ``generate_shuffle_analysis_dataset`` (reference builds it
on Spark :27-49; here the framework's own writer) and
``compute_correlation_distribution`` (:52-85): the abs Pearson
correlation of each shuffled read order against the UNSHUFFLED read
order, sampled over many runs.  A good shuffle concentrates near 0.
"""

import numpy as np

from petastorm_amd.codecs import ScalarCodec
from petastorm_amd.unischema import Unischema, UnischemaField

_ShuffleAnalysisSchema = Unischema('ShuffleAnalysisSchema', [
    UnischemaField('id', np.int64, (), ScalarCodec(), False),
])


def generate_shuffle_analysis_dataset(output_dataset_url, num_rows=1000,
                                      row_group_size=100):
    """id-only dataset sized for shuffle analysis (reference :27-49)."""
    from petastorm_amd.etl.dataset_metadata import materialize_dataset
    with materialize_dataset(output_dataset_url, _ShuffleAnalysisSchema, 1,
                             rows_per_rowgroup=row_group_size) as writer:
        writer.write_rows([{'id': np.int64(i)} for i in range(num_rows)])


def compute_correlation_distribution(dataset_url, id_column,
                                     shuffle_row_drop_partitions=1,
                                     num_corr_samples=10):
    """(mean, std) of abs correlation between shuffled and unshuffled read
    orders over ``num_corr_samples`` runs (reference :52-85)."""
    from petastorm_amd import make_reader

    with make_reader(dataset_url, shuffle_row_groups=False,
                     reader_pool_type='dummy') as reader:
        unshuffled = np.asarray(
            [getattr(row, id_column) for row in reader], dtype=np.float64)

    correlations = []
    for _ in range(num_corr_samples):
        with make_reader(
                dataset_url, shuffle_row_groups=True,
                shuffle_row_drop_partitions=shuffle_row_drop_partitions
                ) as reader:
            shuffled = np.asarray(
                [getattr(row, id_column) for row in reader],
                dtype=np.float64)
        if len(shuffled) < 2 or shuffled.std() == 0 or \
                len(shuffled) != len(unshuffled):
            correlations.append(0.0)
            continue
        correlations.append(
            abs(float(np.corrcoef(unshuffled, shuffled)[0, 1])))
    return float(np.mean(correlations)), float(np.std(correlations))
