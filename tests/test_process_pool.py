"""Process pool end-to-end (spawn is slow; exercised once, not across the
whole e2e matrix)."""
import numpy as np

from petastorm_amd import make_reader


def test_process_pool_roundtrip(test_dataset):
    with make_reader(test_dataset['url'], reader_pool_type='process',
                     workers_count=2, shuffle_row_groups=False) as r:
        rows = list(r)
    assert len(rows) == len(test_dataset['rows'])
    by_id = {int(x.id): x for x in rows}
    src = test_dataset['rows'][0]
    np.testing.assert_array_equal(by_id[int(src['id'])].matrix, src['matrix'])
