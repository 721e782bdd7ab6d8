"""NGram: sliding-window sequence assembly over timestamp-sorted rows.

Parity: /root/reference/petastorm/ngram.py (semantics doc :20-100, assembly
``form_ngram`` :225-270, gap rule ``_ngram_pass_threshold`` :179-193,
per-timestep schema views :215-223, regex field resolution :195-203,
``timestamp_overlap`` :102-125).

Semantics preserved from the reference:

* ``fields`` is a dict mapping a *relative timestep offset* to the list of
  UnischemaFields wanted at that offset.  Offsets need not start at 0; the
  window length is ``max(keys) - min(keys) + 1``.
* Rows inside one row-group are sorted by ``timestamp_field`` and a window
  is emitted at every start position where all consecutive timestamp deltas
  are ``<= delta_threshold``.
* Windows never span row-group boundaries (reference caveat ngram.py:85-91).
* ``timestamp_overlap=False`` makes consecutive emitted windows disjoint in
  timestamps (reference ngram.py:107-117).

MI355X note: on the GPU batch path the same windowing is performed by the
``ngram_window_gather`` HIP kernel over HBM-resident decoded columns; this
class stays the single source of truth for the window rule (the kernel is
validated against :meth:`form_ngram` in tests).
"""

import numpy as np

from petastorm_amd.unischema import Unischema, match_unischema_fields


class NGram(object):
    def __init__(self, fields, delta_threshold, timestamp_field,
                 timestamp_overlap=True):
        """
        :param fields: dict ``{timestep_offset: [UnischemaField or regex str]}``
        :param delta_threshold: max allowed timestamp gap between consecutive
            timesteps in one window
        :param timestamp_field: the UnischemaField (or name) used for ordering
        :param timestamp_overlap: whether consecutive windows may overlap in
            timestamps
        """
        if not isinstance(fields, dict) or not fields:
            raise ValueError('fields must be a non-empty dict of '
                             '{timestep: [fields]}')
        keys = sorted(fields.keys())
        if keys != list(range(min(keys), max(keys) + 1)):
            raise ValueError('NGram timestep keys must be consecutive '
                             'integers; got {}'.format(keys))
        self._fields = {k: list(v) for k, v in fields.items()}
        self.delta_threshold = delta_threshold
        self._timestamp_field = timestamp_field
        self.timestamp_overlap = timestamp_overlap

    # ------------------------------------------------------------------
    @property
    def fields(self):
        return self._fields

    @property
    def timestamp_field(self):
        return self._timestamp_field

    @property
    def timestamp_field_name(self):
        ts = self._timestamp_field
        return ts if isinstance(ts, str) else ts.name

    @property
    def length(self):
        """Window length (reference ngram.py:127-133)."""
        keys = self._fields.keys()
        return max(keys) - min(keys) + 1

    # ------------------------------------------------------------------
    def resolve_regex_field_names(self, schema):
        """Expand any regex-string entries against the full schema
        (reference ngram.py:195-203)."""
        for ts in self._fields:
            self._fields[ts] = match_unischema_fields(schema, self._fields[ts])

    def get_field_names_at_timestep(self, timestep):
        if timestep not in self._fields:
            return []
        return [f.name for f in self._fields[timestep]]

    def get_schema_at_timestep(self, schema, timestep):
        """Schema view of the fields requested at one timestep
        (reference ngram.py:215-223)."""
        return schema.create_schema_view(
            [f for f in self._fields.get(timestep, [])
             if f.name in schema.fields])

    def get_field_names_at_all_timesteps(self):
        names = set()
        for flist in self._fields.values():
            names |= {f.name if not isinstance(f, str) else f for f in flist}
        names.add(self.timestamp_field_name)
        return sorted(names)

    # ------------------------------------------------------------------
    def _window_passes_threshold(self, timestamps):
        """All consecutive deltas <= delta_threshold
        (reference ngram.py:179-193)."""
        for a, b in zip(timestamps[:-1], timestamps[1:]):
            if b - a > self.delta_threshold:
                return False
        return True

    def form_ngram(self, data, schema):
        """Assemble windows from a list of decoded row dicts.

        :param data: list of row dicts (one row-group's rows)
        :param schema: the schema the rows follow
        :return: list of dicts ``{timestep: namedtuple}``
        (reference ngram.py:225-270)
        """
        ts_name = self.timestamp_field_name
        rows = sorted(data, key=lambda r: r[ts_name])
        n = len(rows)
        length = self.length
        base = min(self._fields.keys())
        result = []
        next_start = 0
        for start in range(0, n - length + 1):
            if start < next_start:
                continue
            window = rows[start:start + length]
            if not self._window_passes_threshold([r[ts_name] for r in window]):
                continue
            ngram = {}
            for offset_idx, row in enumerate(window):
                ts_key = base + offset_idx
                view = self.get_schema_at_timestep(schema, ts_key)
                ngram[ts_key] = view.make_namedtuple(
                    **{name: row[name] for name in view.fields})
            result.append(ngram)
            if not self.timestamp_overlap:
                next_start = start + length
        return result

    def form_ngram_indices(self, timestamps):
        """Window *start indices* over a sorted timestamp array.

        This is the pure index computation the GPU ``ngram_window_gather``
        kernel reproduces; kept separate so kernel tests can compare
        index-for-index.
        """
        ts = np.asarray(timestamps)
        order = np.argsort(ts, kind='stable')
        ts_sorted = ts[order]
        length = self.length
        starts = []
        next_start = 0
        for s in range(0, len(ts_sorted) - length + 1):
            if s < next_start:
                continue
            if self._window_passes_threshold(ts_sorted[s:s + length]):
                starts.append(s)
                if not self.timestamp_overlap:
                    next_start = s + length
        return order, np.asarray(starts, dtype=np.int64)

    # ------------------------------------------------------------------
    def make_namedtuple(self, schema, ngram_dict):
        """Wrap a {timestep: dict} into {timestep: namedtuple}
        (reference ngram.py:272-297)."""
        out = {}
        for ts_key, value in ngram_dict.items():
            view = self.get_schema_at_timestep(schema, ts_key)
            out[ts_key] = view.make_namedtuple(**value) \
                if isinstance(value, dict) else value
        return out
