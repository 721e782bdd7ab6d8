"""Spark DataFrame -> training loaders via the dataset converter.

Parity role: /root/reference/examples/spark_dataset_converter/
(pytorch_converter_example.py, tensorflow_converter_example.py): build a
DataFrame, `make_spark_converter` materializes it to Parquet under the
configured cache dir (deduped by logical plan), then
`make_torch_dataloader` / `make_tf_dataset` feed training.

Offline note: pyspark needs a JVM and is not installable here, so when the
real package is absent this example runs against the pandas-backed stand-in
(`petastorm_amd.test_util.pyspark_stub`) — the converter code path is the
same either way.  TensorFlow likewise falls back to the graph/eager
stand-in (`tf_stub`).

Run:  python examples/spark_converter/main.py [--rows N]
"""
import argparse
import os
import sys
import tempfile

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))

import numpy as np
import torch


def get_spark_session():
    try:
        from pyspark.sql import SparkSession  # noqa: F401
        import pyspark
        print('using real pyspark %s' % pyspark.__version__)
        return (pyspark.sql.SparkSession.builder
                .master('local[2]').appName('converter-example')
                .getOrCreate()), False
    except ImportError:
        from petastorm_amd.test_util import pyspark_stub
        mods = pyspark_stub.build_modules()
        sys.modules.update(mods)
        print('pyspark not installed; using the pandas-backed stand-in')
        return pyspark_stub.SparkSession(), True


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument('--rows', type=int, default=200)
    parser.add_argument('--epochs', type=int, default=1)
    args = parser.parse_args()

    from petastorm_amd.spark import (SparkDatasetConverter,
                                     make_spark_converter)

    spark, stubbed = get_spark_session()
    cache_dir = tempfile.mkdtemp(prefix='sdc_cache_')
    spark.conf.set(SparkDatasetConverter.PARENT_CACHE_DIR_URL_CONF,
                   'file://' + cache_dir)

    # a feature matrix as plain columns (works on stub and real Spark)
    import pandas as pd
    pdf = pd.DataFrame({
        'x0': np.random.rand(args.rows).astype(np.float64),
        'x1': np.random.rand(args.rows).astype(np.float64),
        'label': (np.random.rand(args.rows) > 0.5).astype(np.int64),
    })
    df = spark.createDataFrame(pdf)

    # float64 -> float32 narrowing happens inside the converter
    converter = make_spark_converter(df, dtype='float32')
    print('materialized %d rows -> %s' % (len(converter),
                                          converter.file_urls))

    # --- PyTorch route ----------------------------------------------------
    model = torch.nn.Linear(2, 2)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    loss_fn = torch.nn.CrossEntropyLoss()
    steps = 0
    with converter.make_torch_dataloader(batch_size=32,
                                         num_epochs=args.epochs,
                                         shuffle_row_groups=False) as loader:
        for batch in loader:
            x = torch.stack([batch['x0'], batch['x1']], dim=1).float()
            y = batch['label'].long()
            opt.zero_grad()
            loss = loss_fn(model(x), y)
            loss.backward()
            opt.step()
            steps += 1
    print('torch: %d optimizer steps, final loss %.4f' %
          (steps, float(loss)))

    # --- TF route ---------------------------------------------------------
    try:
        import tensorflow  # noqa: F401
        have_tf = True
    except ImportError:
        from petastorm_amd.test_util import tf_stub
        sys.modules['tensorflow'] = tf_stub.build_module()
        have_tf = False
        print('tensorflow not installed; using the tf stand-in')
    batches = 0
    with converter.make_tf_dataset(batch_size=32,
                                   num_epochs=1) as dataset:
        for t in dataset:
            assert hasattr(t, 'label')
            batches += 1
    print('tf%s: %d batches' % ('' if have_tf else '-stub', batches))

    converter.delete()
    print('cache deleted: OK')


if __name__ == '__main__':
    main()
