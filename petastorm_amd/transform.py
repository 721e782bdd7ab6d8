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


class FusedImageNormalize(object):
    """The dominant training-input transform, declared so the GPU route can
    FUSE it into the JPEG color kernel epilogue: uint8 HWC image ->
    ``((u8/`` `scale_div` ``) - mean) / std`` as NCHW float32.

    Use through :func:`fused_image_normalize`.  Works on every route:

    * GPU route + jpeg column: the decoder writes normalized NCHW fp32
      directly (no NHWC uint8 intermediate, no separate normalize kernel).
    * any other route: ``__call__`` performs the same math on the
      column-dict batch (torch on device, numpy on CPU) — the callable
      detects an already-fused column (float32 NCHW) and passes it through.
    """

    def __init__(self, field, mean, std, scale_div=255.0):
        self.field = field
        self.mean = [float(m) for m in mean]
        self.std = [float(s) for s in std]
        self.scale_div = float(scale_div)

    def __call__(self, columns):
        import numpy as _np
        img = columns.get(self.field)
        if img is None:
            return columns
        out = dict(columns)
        try:
            import torch as _torch
            is_torch = isinstance(img, _torch.Tensor)
        except ImportError:
            is_torch = False
        if is_torch:
            import torch as _torch
            if img.dtype == _torch.float32 and img.dim() == 4 and \
                    img.shape[1] == 3:
                return columns  # already fused by the decoder
            mean = _torch.tensor(self.mean, device=img.device)
            std = _torch.tensor(self.std, device=img.device)
            x = img.to(_torch.float32) / self.scale_div
            x = (x - mean) / std
            out[self.field] = x.permute(0, 3, 1, 2).contiguous()
        else:
            arr = _np.asarray(img)
            if arr.dtype == _np.float32 and arr.ndim == 4 and \
                    arr.shape[1] == 3:
                return columns
            single = arr.ndim == 3  # row path: one HWC image
            if single:
                arr = arr[None]
            x = arr.astype(_np.float32) / self.scale_div
            x = (x - _np.asarray(self.mean, dtype=_np.float32)) / \
                _np.asarray(self.std, dtype=_np.float32)
            x = _np.ascontiguousarray(x.transpose(0, 3, 1, 2))
            out[self.field] = x[0] if single else x
        return out


def fused_image_normalize(field, mean, std, scale_div=255.0,
                          extra_edit_fields=None):
    """A TransformSpec normalizing ``field`` (uint8 HWC image) into NCHW
    float32 — fused into the GPU JPEG decode when possible."""
    import numpy as _np
    func = FusedImageNormalize(field, mean, std, scale_div)
    edits = [UnischemaField(field, _np.float32, (3, None, None), None,
                            False)]
    if extra_edit_fields:
        edits.extend(extra_edit_fields)
    return TransformSpec(func, edit_fields=edits)


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


def edit_field(name, numpy_dtype, shape, nullable=False):
    """Helper building one ``edit_fields`` entry (reference
    transform.py:19-25)."""
    return (name, numpy_dtype, shape, nullable)
