import sys
import os

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import pytest  # noqa: E402

from petastorm_amd.test_util.dataset_gen import (  # noqa: E402
    create_scalar_dataset, create_test_dataset)


def pytest_configure(config):
    config.addinivalue_line(
        'markers', 'gpu: test requires an AMD GPU (MI355X); skipped on CPU-only runners')


@pytest.fixture(scope='session')
def test_dataset(tmp_path_factory):
    """TestSchema dataset + ground-truth rows (reference
    tests/conftest.py:52-101 'synthetic_dataset')."""
    path = tmp_path_factory.mktemp('synthetic_dataset')
    url = 'file://' + str(path)
    rows = create_test_dataset(url, num_rows=60, num_files=2,
                               rowgroup_size_mb=0.02, seed=0)
    return {'url': url, 'path': str(path), 'rows': rows}


@pytest.fixture(scope='session')
def scalar_dataset(tmp_path_factory):
    """Plain-parquet store (reference tests/test_common.py:161-245)."""
    path = tmp_path_factory.mktemp('scalar_dataset')
    url = 'file://' + str(path)
    cols = create_scalar_dataset(url, num_rows=500, rowgroup_size=100, seed=1)
    return {'url': url, 'path': str(path), 'cols': cols}
