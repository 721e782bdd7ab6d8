"""Row-group selectors: query-time lookups against inverted row-group indexes.

Parity: /root/reference/petastorm/selectors.py:20-100.

* ``RowGroupSelectorBase``: ``select_row_groups(index_dict) -> set`` of
  row-group ordinals
* ``SingleIndexSelector`` (:32-50), ``IntersectIndexSelector`` (:53-75),
  ``UnionIndexSelector`` (:78-100)

Indexes themselves are built by petastorm_amd.etl.rowgroup_indexing and
stored as a JSON sidecar (not pickled parquet metadata).
"""


class RowGroupSelectorBase(object):
    def select_index_names(self):
        """Names of the indexes this selector needs."""
        raise NotImplementedError()

    def select_row_groups(self, index_dict):
        """:param index_dict: name -> RowGroupIndexerBase with loaded data"""
        raise NotImplementedError()


class SingleIndexSelector(RowGroupSelectorBase):
    """Rows groups containing any of the given values for one index."""

    def __init__(self, index_name, values_list):
        self._index_name = index_name
        self._values = list(values_list)

    def select_index_names(self):
        return [self._index_name]

    def select_row_groups(self, index_dict):
        indexer = index_dict[self._index_name]
        out = set()
        for v in self._values:
            out |= set(indexer.get_row_group_indexes(v))
        return out


class IntersectIndexSelector(RowGroupSelectorBase):
    """Row groups selected by *all* of the given single-index selectors."""

    def __init__(self, single_index_selectors):
        self._selectors = list(single_index_selectors)

    def select_index_names(self):
        names = []
        for s in self._selectors:
            names.extend(s.select_index_names())
        return names

    def select_row_groups(self, index_dict):
        sets = [s.select_row_groups(index_dict) for s in self._selectors]
        return set.intersection(*sets) if sets else set()


class UnionIndexSelector(RowGroupSelectorBase):
    """Row groups selected by *any* of the given single-index selectors."""

    def __init__(self, single_index_selectors):
        self._selectors = list(single_index_selectors)

    def select_index_names(self):
        names = []
        for s in self._selectors:
            names.extend(s.select_index_names())
        return names

    def select_row_groups(self, index_dict):
        out = set()
        for s in self._selectors:
            out |= s.select_row_groups(index_dict)
        return out
