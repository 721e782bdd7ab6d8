"""Execute the Spark dataset converter (spark_dataset_converter) against
the pandas-backed pyspark stand-in: materialization, plan dedupe, float
precision + vector conversion, make_torch_dataloader, make_tf_dataset,
file-wait, shard-consistency warning, delete — the flows the reference
covers in tests/test_spark_dataset_converter.py.  (pyspark/JVM are not
installable offline; see pyspark_stub docstring.)"""
import sys

import numpy as np
import pandas as pd
import pytest

from petastorm_amd.spark import spark_dataset_converter as sdc
from petastorm_amd.test_util import pyspark_stub, tf_stub


@pytest.fixture()
def spark(monkeypatch, tmp_path):
    mods = pyspark_stub.build_modules()
    for name, mod in mods.items():
        monkeypatch.setitem(sys.modules, name, mod)
    session = pyspark_stub.SparkSession()
    session.conf.set(sdc.CACHE_DIR_CONF_KEY,
                     'file://' + str(tmp_path / 'cache'))
    # per-test isolation of the module-level converter cache
    monkeypatch.setattr(sdc, '_converter_cache', {})
    (tmp_path / 'cache').mkdir()
    return session


def _make_df(spark, n=100, source_id='src0'):
    pdf = pd.DataFrame({
        'id': np.arange(n, dtype=np.int64),
        'x': np.linspace(0.0, 1.0, n).astype(np.float64),
        'y': (np.arange(n) % 5).astype(np.int32),
    })
    return spark.createDataFrame(pdf, source_id=source_id)


def test_materialize_and_read_back(spark):
    df = _make_df(spark)
    conv = sdc.make_spark_converter(df)
    assert len(conv) == 100
    assert conv.file_urls
    from petastorm_amd import make_batch_reader
    with make_batch_reader(conv.file_urls, num_epochs=1,
                           shuffle_row_groups=False) as r:
        ids = np.concatenate([np.asarray(b.id) for b in r])
    assert sorted(ids.tolist()) == list(range(100))


def test_float_precision_narrowing(spark):
    df = _make_df(spark)
    conv = sdc.make_spark_converter(df, dtype='float32')
    import pyarrow.parquet as pq
    from petastorm_amd.fs_utils import get_filesystem_and_path_or_paths
    fs, paths = get_filesystem_and_path_or_paths(conv.file_urls)
    sch = pq.ParquetFile(paths[0]).schema_arrow
    assert str(sch.field('x').type) == 'float'  # float64 -> float32


def test_vector_columns_converted(spark):
    n = 20
    pdf = pd.DataFrame({
        'id': np.arange(n, dtype=np.int64),
        'features': [pyspark_stub.DenseVector([i, i + 0.5, i + 1.0])
                     for i in range(n)],
    })
    df = spark.createDataFrame(pdf, source_id='vec')
    conv = sdc.make_spark_converter(df, dtype='float32')
    from petastorm_amd import make_batch_reader
    with make_batch_reader(conv.file_urls, num_epochs=1,
                           shuffle_row_groups=False) as r:
        b = next(iter(r))
    feats = np.stack([np.asarray(v) for v in b.features])
    assert feats.shape == (n, 3) and feats.dtype == np.float32


def test_plan_dedupe_same_result(spark):
    """Identical logical plans must reuse the materialization; different
    plans must not (reference sameResult dedupe :516-524)."""
    df1 = _make_df(spark, source_id='shared')
    df2 = _make_df(spark, source_id='shared')   # same plan
    df3 = _make_df(spark, source_id='other')    # different source
    c1 = sdc.make_spark_converter(df1)
    c2 = sdc.make_spark_converter(df2)
    c3 = sdc.make_spark_converter(df3)
    assert c1 is c2
    assert c3 is not c1


def test_make_torch_dataloader(spark):
    df = _make_df(spark)
    conv = sdc.make_spark_converter(df)
    with conv.make_torch_dataloader(batch_size=16, num_epochs=1) as loader:
        seen = 0
        for batch in loader:
            seen += len(batch['id'])
    assert seen == 100


def test_make_tf_dataset(spark, monkeypatch):
    monkeypatch.setitem(sys.modules, 'tensorflow', tf_stub.build_module())
    df = _make_df(spark)
    conv = sdc.make_spark_converter(df)
    with conv.make_tf_dataset(batch_size=32, num_epochs=1) as ds:
        ids = []
        for batch in ds:
            arr = batch.id.numpy()
            assert arr.shape[0] <= 32
            ids.extend(int(v) for v in arr)
    assert sorted(ids) == list(range(100))


def test_missing_cache_dir_conf_raises(spark):
    spark.conf.set(sdc.CACHE_DIR_CONF_KEY, '')
    df = _make_df(spark)
    with pytest.raises(ValueError):
        sdc.make_spark_converter(df)


def test_wait_file_available_timeout(tmp_path):
    missing = 'file://' + str(tmp_path / 'nope' / 'data.parquet')
    with pytest.raises(RuntimeError):
        sdc._wait_file_available([missing], timeout_s=1)


def test_shard_consistency_warning(spark, monkeypatch):
    monkeypatch.setenv('HOROVOD_RANK', '1')
    monkeypatch.setenv('HOROVOD_SIZE', '4')
    df = _make_df(spark)
    conv = sdc.make_spark_converter(df)
    with pytest.warns(UserWarning):
        with conv.make_torch_dataloader(batch_size=8, num_epochs=1,
                                        cur_shard=0, shard_count=1):
            pass


def test_delete_removes_materialization(spark):
    df = _make_df(spark)
    conv = sdc.make_spark_converter(df)
    from petastorm_amd.fs_utils import get_filesystem_and_path_or_paths
    fs, paths = get_filesystem_and_path_or_paths(conv.file_urls)
    assert all(fs.exists(p) for p in paths)
    conv.delete()
    assert not any(fs.exists(p) for p in paths)


def test_dataset_as_rdd(spark, tmp_path):
    """spark_utils.dataset_as_rdd executes: per-row-group flatMap read,
    codec decode, namedtuple rows (reference spark_utils.py:23-52)."""
    from petastorm_amd.spark_utils import dataset_as_rdd
    from petastorm_amd.test_util.dataset_gen import create_test_dataset
    url = 'file://' + str(tmp_path / 'rdd_ds')
    rows = create_test_dataset(url, num_rows=20, rowgroup_size_mb=1)
    rdd = dataset_as_rdd(url, spark, schema_fields=['id', 'matrix'])
    got = rdd.collect()
    assert rdd.count() == 20
    by_id = {int(r.id): r for r in got}
    for src in rows:
        np.testing.assert_allclose(by_id[int(src['id'])].matrix,
                                   src['matrix'], rtol=1e-6)


def test_dict_to_spark_row(spark):
    """Write-path compat: dict_to_spark_row encodes through the codecs and
    yields an alphabetically-ordered Row (reference unischema.py:359-406)."""
    from petastorm_amd.codecs import NdarrayCodec, ScalarCodec
    from petastorm_amd.unischema import (Unischema, UnischemaField,
                                         dict_to_spark_row)
    schema = Unischema('R', [
        UnischemaField('zz', np.int64, (), ScalarCodec(), False),
        UnischemaField('aa', np.float32, (3,), NdarrayCodec(), False),
        UnischemaField('mm', np.str_, (), ScalarCodec(), True),
    ])
    row = dict_to_spark_row(schema, {'zz': np.int64(7),
                                     'aa': np.ones(3, np.float32)})
    assert list(row.asDict()) == ['aa', 'mm', 'zz']  # alphabetical
    assert row.zz == 7 and row.mm is None
    assert isinstance(row.aa, bytes)  # npy-encoded by the codec
    back = np.load(__import__('io').BytesIO(row.aa))
    np.testing.assert_array_equal(back, np.ones(3, np.float32))


def test_register_delete_dir_handler(spark):
    """Custom delete handlers plug into materialization cleanup
    (reference :102-114)."""
    deleted = []
    sdc.register_delete_dir_handler(lambda url: deleted.append(url))
    try:
        conv = sdc.make_spark_converter(_make_df(spark, source_id='del'))
        conv.delete()
        assert deleted == [conv.cache_dir_url]
    finally:
        sdc.register_delete_dir_handler(None)
