"""Read-only interop with datasets written by UPSTREAM petastorm.

Real petastorm stores carry a *pickled* ``petastorm.unischema.Unischema``
under the ``dataset-toolkit.unischema.v1`` key of the ``_common_metadata``
file (reference etl/dataset_metadata.py:34-35,194-205,356-385), loaded
through a restricted unpickler with legacy package renames (reference
etl/legacy.py:22-79).

This module depickles that metadata WITHOUT importing petastorm (which is
not installed here): reference class paths resolve to local shim classes
during unpickling, and the shim graph is then converted into this
framework's :class:`~petastorm_amd.unischema.Unischema` + codec objects.
The unpickler is restricted: only an explicit allow-list of modules can
supply real classes; ``petastorm.*`` and ``pyspark.*`` never execute any
foreign code because they resolve to inert shims.
"""

import io
import logging
import pickle
import posixpath

import numpy as np

logger = logging.getLogger(__name__)

#: Key-value metadata keys upstream petastorm writes into _common_metadata
#: (reference etl/dataset_metadata.py:34-35).
REFERENCE_UNISCHEMA_KEY = b'dataset-toolkit.unischema.v1'
REFERENCE_ROWGROUPS_KEY = b'dataset-toolkit.num_row_groups.v1'
REFERENCE_INDEX_KEY = b'dataset-toolkit.rowgroups_index.v1'

COMMON_METADATA = '_common_metadata'

# ---------------------------------------------------------------------------
# Shim classes the pickled reference objects reconstruct into.  Pickle builds
# instances via cls(*args) (namedtuples), cls.__new__(cls) + __dict__ update
# (plain objects), or copyreg._reconstructor; the shims accept all three.


class _Shim(object):
    """Inert stand-in: records constructor args and pickled state."""

    def __new__(cls, *args, **kwargs):
        obj = super(_Shim, cls).__new__(cls)
        obj._shim_args = args
        obj._shim_kwargs = kwargs
        return obj

    def __init__(self, *args, **kwargs):  # state lands via __dict__ update
        pass


class _ShimUnischema(_Shim):
    """petastorm.unischema.Unischema — state: _name, _fields OrderedDict."""


class _ShimUnischemaField(tuple):
    """petastorm.unischema.UnischemaField — a 5-slot namedtuple
    (name, numpy_dtype, shape, codec, nullable).  Tuple-based so both
    pickle reconstruction paths work: NEWOBJ passes the five values as
    positional args; protocol-0 copyreg._reconstructor passes the whole
    value tuple as one argument."""

    def __new__(cls, *args):
        if len(args) == 1 and isinstance(args[0], tuple):
            return super(_ShimUnischemaField, cls).__new__(cls, args[0])
        return super(_ShimUnischemaField, cls).__new__(cls, args)

    def values(self):
        vals = list(self)
        while len(vals) < 5:
            vals.append(None if len(vals) < 4 else False)
        return vals[:5]


class _ShimCompressedImageCodec(_Shim):
    """petastorm.codecs.CompressedImageCodec — state: _image_codec ('.png'),
    _quality."""


class _ShimNdarrayCodec(_Shim):
    """petastorm.codecs.NdarrayCodec — stateless."""


class _ShimCompressedNdarrayCodec(_Shim):
    """petastorm.codecs.CompressedNdarrayCodec — stateless (npz format)."""


class _ShimScalarCodec(_Shim):
    """petastorm.codecs.ScalarCodec — state: _spark_type (a shim)."""


class _ShimSparkType(_Shim):
    """Stand-in for any pyspark.sql.types.* class (DecimalType carries
    precision/scale in __dict__)."""
    type_name = '?'


_PETASTORM_SHIMS = {
    'Unischema': _ShimUnischema,
    'UnischemaField': _ShimUnischemaField,
    'CompressedImageCodec': _ShimCompressedImageCodec,
    'NdarrayCodec': _ShimNdarrayCodec,
    'CompressedNdarrayCodec': _ShimCompressedNdarrayCodec,
    'ScalarCodec': _ShimScalarCodec,
}

_spark_type_shim_cache = {}


def _spark_type_shim(name):
    if name not in _spark_type_shim_cache:
        _spark_type_shim_cache[name] = type(
            '_ShimSpark_' + name, (_ShimSparkType,), {'type_name': name})
    return _spark_type_shim_cache[name]


# Symbols that load for real, by package (reference etl/legacy.py:22-31
# minus petastorm/pyspark, which resolve to shims here instead).  Per-NAME
# allow-lists: admitting a whole module would admit e.g. ``builtins.eval``
# and make the "restricted" unpickler a code-execution vector.
_SAFE_BUILTINS = {'object', 'set', 'frozenset', 'list', 'dict', 'tuple',
                  'bytes', 'bytearray', 'str', 'int', 'float', 'complex',
                  'bool', 'slice', 'range', 'type', 'NoneType'}
# numpy surface a pickled Unischema can reference: the dtype machinery
# (numpy.dtype / numpy.core.multiarray reconstruction helpers) and the
# scalar-type CLASSES fields carry as numpy_dtype.
_SAFE_NUMPY = {'dtype', 'ndarray', 'generic', 'number', '_reconstruct',
               'scalar', '_frombuffer', '_DType_reconstruct',
               'bool_', 'object_', 'bytes_', 'str_', 'string_', 'unicode_',
               'void',
               'int8', 'int16', 'int32', 'int64',
               'uint8', 'uint16', 'uint32', 'uint64',
               'float16', 'float32', 'float64', 'complex64', 'complex128',
               'datetime64', 'timedelta64',
               'byte', 'ubyte', 'short', 'ushort', 'intc', 'uintc',
               'intp', 'uintp', 'longlong', 'ulonglong',
               'half', 'single', 'double', 'longdouble'}
_SAFE_GLOBALS = {
    'collections': {'OrderedDict', 'defaultdict'},
    'decimal': {'Decimal'},
    'copyreg': {'_reconstructor', '__newobj__'},
    'copy_reg': {'_reconstructor', '__newobj__'},
    'builtins': _SAFE_BUILTINS,
    '__builtin__': _SAFE_BUILTINS,
    'numpy': _SAFE_NUMPY,
}
# numpy-1.x scalar-type aliases removed by numpy 2.0 — upstream stores
# pickled years ago reference these classes by their old names.
_NUMPY_RENAMES = {'unicode_': 'str_', 'string_': 'bytes_',
                  'float_': 'float64', 'bool8': 'bool_',
                  'object0': 'object_', 'str0': 'str_', 'bytes0': 'bytes_',
                  'void0': 'void', 'int0': 'intp', 'uint0': 'uintp'}


class RestrictedInteropUnpickler(pickle.Unpickler):
    """Allow-list unpickler mapping reference petastorm/pyspark classes to
    local shims (reference etl/legacy.py:34-48 contract)."""

    def find_class(self, module, name):
        package = module.split('.')[0]
        if package == 'petastorm':
            if name in _PETASTORM_SHIMS:
                return _PETASTORM_SHIMS[name]
            raise pickle.UnpicklingError(
                'unsupported petastorm symbol %s.%s in pickled unischema'
                % (module, name))
        if package == 'pyspark':
            return _spark_type_shim(name)
        if package == 'numpy' and name in _NUMPY_RENAMES \
                and not hasattr(np, name):
            return getattr(np, _NUMPY_RENAMES[name])
        if name in _SAFE_GLOBALS.get(package, ()):
            return super(RestrictedInteropUnpickler, self).find_class(
                module, name)
        raise pickle.UnpicklingError(
            "global '%s.%s' is forbidden" % (module, name))


def _apply_legacy_renames(blob):
    """Rewrite pre-rename package paths inside the pickle stream, exactly as
    the reference does (etl/legacy.py:54-79): the stream-level GLOBAL opcode
    text is the compatibility contract."""
    legacy_packages = ['av.experimental.deepdrive.dataset_toolkit',
                       'av.ml.dataset_toolkit']
    legacy_modules = ['codecs', 'unischema', 'sequence']
    for pkg in legacy_packages:
        for mod in legacy_modules:
            old = '\n(c{}.{}\n'.format(pkg, mod).encode('ascii')
            new = '\n(cpetastorm.{}\n'.format(mod).encode('ascii')
            if old in blob:
                logger.warning(
                    'Depickling legacy "%s.%s" metadata (moved to '
                    'petastorm.%s upstream).', pkg, mod, mod)
                blob = blob.replace(old, new)
    return blob


def restricted_loads(blob):
    return RestrictedInteropUnpickler(io.BytesIO(blob)).load()


# ---------------------------------------------------------------------------
# Shim graph -> this framework's schema objects


def _convert_codec(shim, field_name):
    from petastorm_amd.codecs import (CompressedImageCodec,
                                      CompressedNdarrayCodec, NdarrayCodec,
                                      ScalarCodec)
    if shim is None:
        return None
    if isinstance(shim, _ShimCompressedImageCodec):
        state = shim.__dict__
        image_codec = state.get('_image_codec', '.png').lstrip('.')
        if image_codec == 'jpg':
            image_codec = 'jpeg'
        return CompressedImageCodec(image_codec,
                                    quality=state.get('_quality', 80))
    if isinstance(shim, _ShimCompressedNdarrayCodec):
        return CompressedNdarrayCodec()
    if isinstance(shim, _ShimNdarrayCodec):
        return NdarrayCodec()
    if isinstance(shim, _ShimScalarCodec):
        # The spark type only matters on the reference's write path; this
        # framework derives storage types from the field's numpy dtype.
        return ScalarCodec()
    raise ValueError('Field {!r}: cannot convert pickled codec {!r}'
                     .format(field_name, type(shim).__name__))


def _convert_field(shim):
    from petastorm_amd.unischema import UnischemaField
    name, numpy_dtype, shape, codec, nullable = shim.values()
    if shape is None:
        shape = ()
    return UnischemaField(name, numpy_dtype, tuple(shape),
                          _convert_codec(codec, name), bool(nullable))


def convert_reference_unischema(shim_schema):
    """Pickled reference Unischema (as a shim graph) -> this framework's
    Unischema."""
    from petastorm_amd.unischema import Unischema
    state = shim_schema.__dict__
    name = state.get('_name', 'imported')
    fields_od = state.get('_fields', {})
    fields = [_convert_field(f) for f in fields_od.values()]
    return Unischema(name, fields)


# ---------------------------------------------------------------------------
# Dataset-level entry points


def read_common_metadata_kv(fs, dataset_path):
    """Key-value metadata dict of <dataset>/_common_metadata, or None."""
    import pyarrow.parquet as pq
    if not fs.isdir(dataset_path):
        dataset_path = posixpath.dirname(dataset_path)
    candidate = posixpath.join(dataset_path, COMMON_METADATA)
    if not fs.exists(candidate):
        return None
    with fs.open(candidate, 'rb') as f:
        md = pq.read_metadata(f).metadata
    return md or {}


def load_reference_unischema(fs, path_or_paths):
    """Load an upstream-petastorm pickled Unischema from _common_metadata.

    Returns this framework's Unischema, or None when the store carries no
    reference metadata (not a petastorm dataset).
    Raises on a present-but-undecodable schema — silent fallback to arrow
    inference would return codec fields as raw bytes.
    """
    paths = path_or_paths if isinstance(path_or_paths, list) \
        else [path_or_paths]
    for p in paths:
        kv = read_common_metadata_kv(fs, p)
        if kv is None or REFERENCE_UNISCHEMA_KEY not in kv:
            continue
        blob = kv[REFERENCE_UNISCHEMA_KEY]
        if isinstance(blob, str):
            blob = blob.encode('latin-1')
        shim = restricted_loads(_apply_legacy_renames(blob))
        schema = convert_reference_unischema(shim)
        logger.info('Loaded upstream-petastorm pickled Unischema %r (%d '
                    'fields) from %s/_common_metadata', schema._name,
                    len(schema.fields), p)
        return schema
    return None


def load_reference_rowgroup_counts(fs, dataset_path):
    """num-row-groups-per-file JSON upstream petastorm stores next to the
    schema (reference etl/dataset_metadata.py:208-241), or None."""
    import json
    kv = read_common_metadata_kv(fs, dataset_path)
    if not kv or REFERENCE_ROWGROUPS_KEY not in kv:
        return None
    raw = kv[REFERENCE_ROWGROUPS_KEY]
    if isinstance(raw, bytes):
        raw = raw.decode('utf-8')
    return json.loads(raw)
