"""Unischema: a single schema definition usable across numpy / Parquet / PyTorch.

A :class:`Unischema` extends Parquet's scalar type system with tensor
(ndarray) and compressed-image fields by attaching a :mod:`petastorm_amd.codecs`
codec to each field.

Behavioral parity with the reference (file:line cite into /root/reference):

* ``UnischemaField(name, numpy_dtype, shape, codec, nullable)``
  (petastorm/unischema.py:50-85)
* ``Unischema`` with attribute access per field, cached namedtuple view
  (petastorm/unischema.py:88-111, 174-356)
* ``create_schema_view`` accepting fields or regex patterns
  (petastorm/unischema.py:199-240)
* ``from_arrow_schema`` inference for plain (non-petastorm) Parquet stores
  (petastorm/unischema.py:302-353, 467-502)
* ``insert_explicit_nulls`` (petastorm/unischema.py:409-424)
* ``match_unischema_fields`` (petastorm/unischema.py:437-464)
* row encoding for the write path — reference's ``dict_to_spark_row``
  (petastorm/unischema.py:359-406) becomes :func:`dict_to_encoded_row`
  since this framework writes Parquet through pyarrow, not Spark.

Differences by design (MI355X-first, not a port):

* The schema is serialized as JSON (:func:`Unischema.to_json` /
  :func:`Unischema.from_json`) instead of pickle — the reference acknowledges
  pickle as a fragility (petastorm/etl/dataset_metadata.py:202-204).
* No >255-field namedtuple workaround is needed on Python >= 3.7
  (reference petastorm/namedtuple_gt_255_fields.py existed for older
  interpreters); we simply build the namedtuple directly.
"""

import copy
import re
import json
import warnings
from collections import OrderedDict, namedtuple
from decimal import Decimal

import numpy as np

_NUMPY_TO_ARROW = None  # lazily built (pyarrow import deferred)


def _lazy_pyarrow():
    import pyarrow as pa  # local import: pyarrow is only needed for IO paths
    return pa


class UnischemaField(object):
    """A single field of a :class:`Unischema`.

    :param name: field name (valid python identifier)
    :param numpy_dtype: numpy dtype of the *decoded* in-memory value
        (e.g. ``np.uint8``, ``np.float32``, ``Decimal``, ``np.str_``,
        ``np.bytes_``)
    :param shape: tuple of ints or ``None`` for unknown dimensions. ``()``
        denotes a scalar.
    :param codec: a :class:`petastorm_amd.codecs.DataframeColumnCodec`
        instance or ``None``.  ``None`` means "default codec for this
        dtype/shape" (scalar passthrough for ``()``, ndarray codec
        otherwise), mirroring reference petastorm/unischema.py:69-76.
    :param nullable: whether nulls are permitted.
    """

    __slots__ = ('name', 'numpy_dtype', 'shape', 'codec', 'nullable')

    def __init__(self, name, numpy_dtype, shape=(), codec=None, nullable=False):
        self.name = name
        self.numpy_dtype = numpy_dtype
        self.shape = tuple(shape) if shape is not None else None
        self.codec = codec
        self.nullable = nullable

    # -- value semantics (fields are compared/hased in tests & schema views) --
    def _key(self):
        return (self.name, self.numpy_dtype, self.shape,
                type(self.codec).__name__ if self.codec is not None else None,
                self.nullable)

    def __eq__(self, other):
        if not isinstance(other, UnischemaField):
            return NotImplemented
        return self._key() == other._key()

    def __ne__(self, other):
        return not self == other

    def __hash__(self):
        return hash(self._key())

    def __repr__(self):
        return ('UnischemaField(name={!r}, numpy_dtype={}, shape={}, codec={}, '
                'nullable={})'.format(self.name, getattr(self.numpy_dtype, '__name__', self.numpy_dtype),
                                      self.shape, type(self.codec).__name__ if self.codec else None,
                                      self.nullable))

    # -- serialization ------------------------------------------------------
    def to_dict(self):
        from petastorm_amd import codecs as _codecs
        return {
            'name': self.name,
            'numpy_dtype': _dtype_to_str(self.numpy_dtype),
            'shape': list(self.shape) if self.shape is not None else None,
            'codec': _codecs.codec_to_dict(self.codec),
            'nullable': self.nullable,
        }

    @classmethod
    def from_dict(cls, d):
        from petastorm_amd import codecs as _codecs
        shape = tuple(d['shape']) if d['shape'] is not None else None
        return cls(d['name'], _dtype_from_str(d['numpy_dtype']), shape,
                   _codecs.codec_from_dict(d['codec']), d['nullable'])


_SPECIAL_DTYPES = {
    'Decimal': Decimal,
    'str_': np.str_,
    'unicode_': np.str_,
    'bytes_': np.bytes_,
    'string_': np.bytes_,
    'object_': np.object_,
    'datetime64': np.datetime64,
    'bool_': np.bool_,
}


def _dtype_to_str(dt):
    if dt is Decimal:
        return 'Decimal'
    if dt is np.str_:
        return 'str_'
    if dt is np.bytes_:
        return 'bytes_'
    if dt is np.object_:
        return 'object_'
    if dt is np.datetime64:
        return 'datetime64'
    if isinstance(dt, np.dtype):
        return dt.name
    return np.dtype(dt).name


def _dtype_from_str(s):
    if s in _SPECIAL_DTYPES:
        return _SPECIAL_DTYPES[s]
    return np.dtype(s).type


class _NamedtupleCache(object):
    """Cache of generated namedtuple types keyed by (schema name, fields).

    Reference: petastorm/unischema.py:88-111.  Two Unischema instances with
    identical name+field-names share one namedtuple class so equality of rows
    read by different readers holds.
    """

    _store = {}

    @classmethod
    def get(cls, parent_name, field_names):
        key = (parent_name, tuple(field_names))
        if key not in cls._store:
            cls._store[key] = namedtuple(parent_name, list(field_names))
        return cls._store[key]


class Unischema(object):
    """An ordered collection of :class:`UnischemaField` with named access.

    ``schema.fields`` is an ``OrderedDict`` name->field; each field is also an
    attribute (``schema.my_field``).  Reference: petastorm/unischema.py:174-356.
    """

    def __init__(self, name, fields):
        self._name = name
        # Sort alphabetically to give a deterministic field order, matching
        # the reference's default behavior (petastorm/unischema.py:36,
        # _UNISCHEMA_FIELD_ORDER = 'alphabetical').
        self._fields = OrderedDict(sorted(((f.name, f) for f in fields),
                                          key=lambda t: t[0]))
        for f in self._fields.values():
            setattr(self, f.name, f)

    # ------------------------------------------------------------------
    @property
    def fields(self):
        return self._fields

    def __getattr__(self, item):
        # NB: guard against recursion during unpickling, when instance
        # attributes are not yet restored
        if item.startswith('_'):
            raise AttributeError(item)
        name = self.__dict__.get('_name', '<unnamed>')
        raise AttributeError('Unischema {!r} has no field {!r}'.format(
            name, item))

    def create_schema_view(self, fields):
        """Return a new Unischema with a subset of the fields.

        ``fields`` may contain :class:`UnischemaField` instances and/or
        strings, where strings are treated as regular expressions fully
        matched against field names (reference petastorm/unischema.py:199-240).
        """
        fields = list(fields)
        for f in fields:
            if isinstance(f, UnischemaField):
                if f.name not in self._fields:
                    raise ValueError('field {} does not belong to the schema {}'
                                     .format(f, self._name))
        matched = match_unischema_fields(self, fields)
        # keep original declaration order within the view
        view_fields = [f for f in self._fields.values() if f in matched]
        return Unischema('{}_view'.format(self._name), view_fields)

    def make_namedtuple(self, **kwargs):
        """Build one row instance of this schema's namedtuple type.

        Missing nullable fields become None (reference
        petastorm/unischema.py:330-344 make_namedtuple semantics: every field
        must be provided; we keep that strictness).
        """
        typ = self._get_namedtuple()
        return typ(**{k: kwargs[k] for k in typ._fields})

    def make_namedtuple_from_dict(self, row_dict):
        typ = self._get_namedtuple()
        return typ(**{k: row_dict[k] for k in typ._fields})

    def _get_namedtuple(self):
        return _NamedtupleCache.get(self._name, list(self._fields.keys()))

    def __repr__(self):
        lines = ['Unischema({},'.format(self._name)]
        lines += ['  {!r},'.format(f) for f in self._fields.values()]
        lines.append(')')
        return '\n'.join(lines)

    # ------------------------------------------------------------------
    # Arrow interop
    # ------------------------------------------------------------------
    def as_arrow_schema(self):
        """Render the *storage* schema as a pyarrow schema (for the writer)."""
        pa = _lazy_pyarrow()
        from petastorm_amd import codecs as _codecs
        pa_fields = []
        for f in self._fields.values():
            codec = _codecs.effective_codec(f)
            pa_fields.append(pa.field(f.name, codec.arrow_type(f), f.nullable))
        return pa.schema(pa_fields)

    @classmethod
    def from_arrow_schema(cls, arrow_schema, omit_unsupported_fields=True,
                          name='inferred_schema'):
        """Infer a Unischema from a plain (non-petastorm) Arrow schema.

        Reference: petastorm/unischema.py:302-353 and the arrow-type->numpy
        mapping at petastorm/unischema.py:467-502.
        """
        fields = []
        for i in range(len(arrow_schema.names)):
            field = arrow_schema.field(i)
            try:
                np_dtype, shape, codec = _numpy_and_codec_from_arrow_type(field.type)
            except ValueError:
                if omit_unsupported_fields:
                    warnings.warn('Column {!r} has an unsupported arrow type {} '
                                  'and is omitted from the inferred schema'
                                  .format(field.name, field.type))
                    continue
                raise
            fields.append(UnischemaField(field.name, np_dtype, shape, codec,
                                         field.nullable))
        return cls(name, fields)

    # ------------------------------------------------------------------
    # JSON serialization (replaces the reference's pickled schema,
    # petastorm/etl/dataset_metadata.py:194-205)
    # ------------------------------------------------------------------
    def to_json(self):
        return json.dumps({
            'version': 1,
            'name': self._name,
            'fields': [f.to_dict() for f in self._fields.values()],
        })

    @classmethod
    def from_json(cls, s):
        d = json.loads(s)
        return cls(d['name'], [UnischemaField.from_dict(fd) for fd in d['fields']])


def _numpy_and_codec_from_arrow_type(arrow_type):
    """Map a pyarrow DataType to (numpy_dtype, shape, codec).

    Mirrors reference petastorm/unischema.py:467-502: scalar arrow types map
    to scalar fields; list<primitive> maps to a 1-D ndarray field; binary maps
    to bytes; string to unicode; decimal to Decimal.
    """
    pa = _lazy_pyarrow()
    t = arrow_type
    import pyarrow.types as pt
    if pt.is_int8(t):
        return np.int8, (), None
    if pt.is_uint8(t):
        return np.uint8, (), None
    if pt.is_int16(t):
        return np.int16, (), None
    if pt.is_uint16(t):
        return np.uint16, (), None
    if pt.is_int32(t):
        return np.int32, (), None
    if pt.is_uint32(t):
        return np.uint32, (), None
    if pt.is_int64(t):
        return np.int64, (), None
    if pt.is_uint64(t):
        return np.uint64, (), None
    if pt.is_float32(t):
        return np.float32, (), None
    if pt.is_float64(t):
        return np.float64, (), None
    if pt.is_boolean(t):
        return np.bool_, (), None
    if pt.is_float16(t):
        return np.float16, (), None
    if pt.is_string(t) or pt.is_large_string(t):
        return np.str_, (), None
    if pt.is_binary(t) or pt.is_large_binary(t) or \
            pt.is_fixed_size_binary(t):
        # fixed_size_binary -> bytes, like the reference's np.string_
        # mapping (unischema.py:491-493)
        return np.bytes_, (), None
    if pt.is_decimal(t):
        return Decimal, (), None
    if pt.is_date(t) or pt.is_timestamp(t):
        return np.datetime64, (), None
    if pt.is_list(t) or pt.is_large_list(t) or pt.is_fixed_size_list(t):
        sub_dtype, sub_shape, _ = _numpy_and_codec_from_arrow_type(t.value_type)
        if sub_shape != ():
            raise ValueError('Nested lists are not supported: {}'.format(t))
        n = t.list_size if pt.is_fixed_size_list(t) else None
        return sub_dtype, (n,), None
    raise ValueError('Unsupported arrow type: {}'.format(t))


def match_unischema_fields(schema, field_list):
    """Resolve a mixed list of UnischemaFields and regex strings.

    Strings are *fully matched* (``re.fullmatch`` semantics) against field
    names; reference petastorm/unischema.py:437-464 (which uses
    ``re.match`` anchored patterns — documented there to full-match since
    0.13).  Returns the list of matched UnischemaField objects.
    """
    if field_list is None:
        return list(schema.fields.values())
    matched = []
    for item in field_list:
        if isinstance(item, UnischemaField):
            matched.append(item)
        elif isinstance(item, str):
            pat = re.compile(item)
            matched.extend(f for f in schema.fields.values()
                           if pat.fullmatch(f.name))
        else:
            raise ValueError('Elements of the field list must be '
                             'UnischemaField or string (regex); got {!r}'
                             .format(item))
    # de-dup preserving order
    seen, out = set(), []
    for f in matched:
        if f.name not in seen:
            seen.add(f.name)
            out.append(f)
    return out


def insert_explicit_nulls(unischema, row_dict):
    """Insert ``None`` for missing nullable fields; raise on missing non-nullable.

    Reference: petastorm/unischema.py:409-424.
    """
    for name, field in unischema.fields.items():
        if name not in row_dict:
            if field.nullable:
                row_dict[name] = None
            else:
                raise ValueError('Field {} is not found in the row_dict, but '
                                 'is not nullable.'.format(name))


def dict_to_encoded_row(unischema, row_dict):
    """Encode a row dict into Parquet-storable values using field codecs.

    The reference's write-path equivalent is ``dict_to_spark_row``
    (petastorm/unischema.py:359-406).  Verifies the key set matches the
    schema, applies each field's codec, and inserts explicit nulls.
    """
    from petastorm_amd import codecs as _codecs
    if not isinstance(row_dict, dict):
        raise TypeError('row_dict must be a dict, got {}'.format(type(row_dict)))
    row = copy.copy(row_dict)
    insert_explicit_nulls(unischema, row)
    if set(row.keys()) != set(unischema.fields.keys()):
        raise ValueError('Dictionary fields {} do not match schema fields {}'
                         .format(sorted(row.keys()), sorted(unischema.fields.keys())))
    encoded = {}
    for name, value in row.items():
        field = unischema.fields[name]
        if value is None:
            if not field.nullable:
                raise ValueError('Field {} is not nullable but got None'.format(name))
            encoded[name] = None
        else:
            encoded[name] = _codecs.effective_codec(field).encode(field, value)
    return encoded


def dict_to_spark_row(unischema, row_dict):
    """Encode a row dict into a ``pyspark.sql.Row`` for Spark-side dataset
    writes (reference unischema.py:359-406: codecs encode each field,
    explicit nulls are inserted, and Row fields sort alphabetically —
    Spark matches schema to Row by POSITION, reference's warning at
    :225-227).  Requires pyspark."""
    from collections import OrderedDict

    from pyspark.sql import Row
    encoded = dict_to_encoded_row(unischema, row_dict)
    return Row(**OrderedDict(sorted(encoded.items(), key=lambda kv: kv[0])))
