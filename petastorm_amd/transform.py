"""TransformSpec: user transform applied inside the decode workers, with a
declarative post-transform schema mutation.

Parity: /root/reference/petastorm/transform.py:27-89.

* ``TransformSpec(func, edit_fields, removed_fields, selected_fields)``
* ``transform_schema(schema, transform_spec)`` produces the schema the
  reader's output rows follow after the transform ran.

In the row path the func receives a row dict; in the batch path it receives
a dict of column arrays (this framework is pandas-free on the hot path — the
reference passes a DataFrame, see petastorm/arrow_reader_worker.py:247-277;
a column-dict is the columnar equivalent).  GPU-resident batches expose the
same dict interface with torch tensors.
"""

from petastorm_amd.unischema import Unischema, UnischemaField


class TransformSpec(object):
    """Declares a transform function and its effect on the schema.

    :param func: callable applied to each row dict (row path) or column-dict
        batch (batch path).  May be ``None`` when only field
        removal/selection is desired.
    :param edit_fields: list of ``UnischemaField`` or 4/5-tuples
        ``(name, numpy_dtype, shape, is_nullable)`` describing fields the
        transform adds or modifies (reference transform.py:38-47).
    :param removed_fields: list of field names removed by the transform.
    :param selected_fields: if not None, the exact output field name list
        (applied after edits/removals).
    """

    def __init__(self, func=None, edit_fields=None, removed_fields=None,
                 selected_fields=None):
        self.func = func
        self.edit_fields = edit_fields or []
        self.removed_fields = removed_fields or []
        self.selected_fields = selected_fields

    def __eq__(self, other):
        return isinstance(other, TransformSpec) and self.__dict__ == other.__dict__


def _as_unischema_field(entry):
    if isinstance(entry, UnischemaField):
        return entry
    if isinstance(entry, (tuple, list)):
        if len(entry) == 4:
            name, np_dtype, shape, nullable = entry
            return UnischemaField(name, np_dtype, shape, None, nullable)
        if len(entry) == 5:
            name, np_dtype, shape, codec, nullable = entry
            return UnischemaField(name, np_dtype, shape, codec, nullable)
    raise ValueError('edit_fields entries must be UnischemaField or '
                     '(name, numpy_dtype, shape, nullable) tuples; got {!r}'
                     .format(entry))


def transform_schema(schema, transform_spec):
    """Apply a TransformSpec's schema mutation (reference transform.py:60-89)."""
    fields = dict(schema.fields)
    for entry in transform_spec.edit_fields:
        f = _as_unischema_field(entry)
        fields[f.name] = f
    for name in transform_spec.removed_fields:
        fields.pop(name, None)
    if transform_spec.selected_fields is not None:
        unknown = set(transform_spec.selected_fields) - set(fields)
        if unknown:
            raise ValueError('selected_fields contains unknown fields: {}'
                             .format(sorted(unknown)))
        fields = {k: v for k, v in fields.items()
                  if k in set(transform_spec.selected_fields)}
    return Unischema(schema._name + '_transformed', list(fields.values()))
