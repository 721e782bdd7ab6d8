"""Spark integration (lazy: pyspark is imported only when used).

Parity: /root/reference/petastorm/spark/__init__.py:16.
"""

from petastorm_amd.spark.spark_dataset_converter import (  # noqa: F401
    SparkDatasetConverter, make_spark_converter)
