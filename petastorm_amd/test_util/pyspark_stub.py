"""A minimal pyspark stand-in for executing the Spark dataset converter.

pyspark (and a JVM) are not installable in this offline environment, so
``spark_dataset_converter`` runs against this stub: a pandas-backed
DataFrame implementing exactly the surface the converter touches —
``df.schema`` fields with Spark types, ``withColumn``+``cast`` for the
float-precision pass, ML ``VectorUDT`` + ``vector_to_array``, logical-plan
``semanticHash`` for the plan-equality dedupe (reference
spark_dataset_converter.py:516-524), ``count``, and
``df.write.option(...).parquet(url)`` which writes real Parquet through
pyarrow.  A TEST HARNESS, not a Spark implementation.
"""

import posixpath
import types

import numpy as np


# --- pyspark.sql.types -----------------------------------------------------

class DataType(object):
    def __repr__(self):
        return type(self).__name__


class FloatType(DataType):
    pass


class DoubleType(DataType):
    pass


class IntegerType(DataType):
    pass


class LongType(DataType):
    pass


class StringType(DataType):
    pass


class VectorUDT(DataType):
    pass


class StructField(object):
    def __init__(self, name, dataType):
        self.name = name
        self.dataType = dataType


class DenseVector(object):
    def __init__(self, values):
        self.values = np.asarray(values, dtype=np.float64)


def _infer_type(series):
    if len(series) and isinstance(series.iloc[0], DenseVector):
        return VectorUDT()
    kind = series.dtype
    if kind == np.float32:
        return FloatType()
    if kind == np.float64:
        return DoubleType()
    if kind == np.int32:
        return IntegerType()
    if kind == np.int64:
        return LongType()
    return StringType()


# --- column expressions (enough for col(name).cast(type)) ------------------

class Column(object):
    def __init__(self, name, cast_to=None):
        self.name = name
        self.cast_to = cast_to

    def cast(self, target):
        return Column(self.name, cast_to=target)


def col(name):
    return Column(name)


_SPARK_TO_NUMPY = {FloatType: np.float32, DoubleType: np.float64,
                   IntegerType: np.int32, LongType: np.int64}


# --- DataFrame / session ---------------------------------------------------

class _Writer(object):
    def __init__(self, df):
        self._df = df
        self._options = {}

    def option(self, k, v):
        self._options[k] = v
        return self

    def parquet(self, url):
        import pyarrow as pa
        import pyarrow.parquet as pq
        from petastorm_amd.fs_utils import get_filesystem_and_path_or_paths
        fs, path = get_filesystem_and_path_or_paths(url)
        fs.makedirs(path, exist_ok=True)
        pdf = self._df._pdf
        cols = {}
        for name in pdf.columns:
            s = pdf[name]
            if len(s) and isinstance(s.iloc[0], (list, np.ndarray)):
                elem = np.asarray(s.iloc[0])
                cols[name] = pa.array(
                    [np.asarray(v) for v in s],
                    type=pa.list_(pa.from_numpy_dtype(elem.dtype)))
            else:
                cols[name] = pa.array(s.to_numpy())
        table = pa.table(cols)
        comp = self._options.get('compression', 'uncompressed')
        if comp == 'uncompressed':
            comp = 'none'
        block = int(self._options.get('parquet.block.size', 32 << 20))
        # spark's parquet.block.size is bytes per row group; approximate
        # rows/rowgroup from the in-memory size
        nbytes = max(1, int(table.nbytes))
        rows_per_group = max(1, int(len(pdf) * block / nbytes))
        pq.write_table(table,
                       posixpath.join(path, 'part-00000-stub.parquet'),
                       compression=comp, row_group_size=rows_per_group)


class _QueryExecution(object):
    def __init__(self, plan):
        self._plan = plan

    def analyzed(self):
        return self

    def semanticHash(self):
        return hash(self._plan)


class _JDF(object):
    def __init__(self, plan):
        self._plan = plan

    def queryExecution(self):
        return _QueryExecution(self._plan)


class DataFrame(object):
    """pandas-backed; ``plan`` is a hashable description of (source, ops)
    so identical pipelines share a semanticHash like Spark's sameResult."""

    def __init__(self, pdf, session, plan):
        self._pdf = pdf
        self.sparkSession = session
        self._plan = plan
        self._jdf = _JDF(plan)

    @property
    def schema(self):
        return [StructField(c, _infer_type(self._pdf[c]))
                for c in self._pdf.columns]

    def withColumn(self, name, expr):
        pdf = self._pdf.copy()
        if isinstance(expr, Column) and expr.cast_to is not None:
            np_t = _SPARK_TO_NUMPY[type(expr.cast_to)]
            pdf[name] = pdf[expr.name].astype(np_t)
            op = ('cast', name, type(expr.cast_to).__name__)
        elif isinstance(expr, _VectorToArray):
            np_t = np.float32 if expr.dtype == 'float32' else np.float64
            pdf[name] = pdf[expr.name].map(
                lambda v: v.values.astype(np_t))
            op = ('vec2arr', name, expr.dtype)
        else:
            raise TypeError('stub withColumn: unsupported expr')
        return DataFrame(pdf, self.sparkSession, self._plan + (op,))

    def count(self):
        return len(self._pdf)

    def __getitem__(self, name):
        return Column(name)

    @property
    def write(self):
        return _Writer(self)


class _Conf(object):
    def __init__(self):
        self._d = {}

    def set(self, k, v):
        self._d[k] = v

    def get(self, k, default=None):
        return self._d.get(k, default)


class RDD(object):
    """Eager local-list RDD: enough surface for dataset_as_rdd."""

    def __init__(self, items):
        self._items = list(items)

    def flatMap(self, fn):
        return RDD(x for it in self._items for x in fn(it))

    def map(self, fn):
        return RDD(fn(it) for it in self._items)

    def collect(self):
        return list(self._items)

    def count(self):
        return len(self._items)

    def take(self, n):
        return self._items[:n]


class SparkContext(object):
    def parallelize(self, seq, numSlices=None):
        return RDD(seq)


class SparkSession(object):
    def __init__(self):
        self.conf = _Conf()
        self._next_id = 0
        self.sparkContext = SparkContext()

    def createDataFrame(self, pdf, source_id=None):
        if source_id is None:
            source_id = 'df-%d' % self._next_id
            self._next_id += 1
        return DataFrame(pdf, self, (source_id,))


# --- pyspark.ml.functions ---------------------------------------------------

class _VectorToArray(object):
    def __init__(self, column, dtype):
        self.name = column.name
        self.dtype = dtype


def vector_to_array(column, dtype='float64'):
    return _VectorToArray(column, dtype)


class Row(dict):
    """Minimal pyspark.sql.Row: keyword construction, attribute access,
    alphabetical field order preserved by the caller."""

    def __init__(self, **kwargs):
        super(Row, self).__init__(kwargs)
        self.__dict__.update(kwargs)

    def asDict(self):
        return dict(self)


def build_modules():
    """Return {module_name: module} shaped like the pyspark package tree."""
    pyspark = types.ModuleType('pyspark')
    sql = types.ModuleType('pyspark.sql')
    sql_types = types.ModuleType('pyspark.sql.types')
    sql_functions = types.ModuleType('pyspark.sql.functions')
    ml = types.ModuleType('pyspark.ml')
    ml_linalg = types.ModuleType('pyspark.ml.linalg')
    ml_functions = types.ModuleType('pyspark.ml.functions')

    for cls in (FloatType, DoubleType, IntegerType, LongType, StringType,
                StructField):
        setattr(sql_types, cls.__name__, cls)
    sql_functions.col = col
    ml_linalg.VectorUDT = VectorUDT
    ml_linalg.DenseVector = DenseVector
    ml_functions.vector_to_array = vector_to_array

    sql.Row = Row
    pyspark.sql = sql
    sql.types = sql_types
    sql.functions = sql_functions
    pyspark.ml = ml
    ml.linalg = ml_linalg
    ml.functions = ml_functions
    pyspark.SparkSession = SparkSession
    return {
        'pyspark': pyspark,
        'pyspark.sql': sql,
        'pyspark.sql.types': sql_types,
        'pyspark.sql.functions': sql_functions,
        'pyspark.ml': ml,
        'pyspark.ml.linalg': ml_linalg,
        'pyspark.ml.functions': ml_functions,
    }
