"""Predicates (parity: reference tests/test_predicates.py)."""
import numpy as np
import pytest

from petastorm_amd.predicates import (in_intersection, in_lambda, in_negate,
                                      in_pseudorandom_split, in_reduce,
                                      in_set)


def test_in_set():
    p = in_set({1, 2}, 'a')
    assert p.get_fields() == {'a'}
    assert p.do_include({'a': 1}) and not p.do_include({'a': 3})
    mask = p.do_include_vectorized({'a': np.array([1, 2, 3, 4])})
    np.testing.assert_array_equal(mask, [True, True, False, False])


def test_in_intersection():
    p = in_intersection({5, 6}, 'arr')
    assert p.do_include({'arr': np.array([1, 5])})
    assert not p.do_include({'arr': np.array([1, 2])})


def test_in_lambda():
    p = in_lambda(['x', 'y'], lambda v: v['x'] > v['y'])
    assert p.get_fields() == {'x', 'y'}
    assert p.do_include({'x': 2, 'y': 1})
    assert not p.do_include({'x': 1, 'y': 2})


def test_in_lambda_with_state():
    state = {'count': 0}

    def fn(v, s):
        s['count'] += 1
        return v['x'] == 0

    p = in_lambda(['x'], fn, state)
    p.do_include({'x': 0})
    assert state['count'] == 1


def test_in_negate():
    p = in_negate(in_set({1}, 'a'))
    assert not p.do_include({'a': 1}) and p.do_include({'a': 2})
    mask = p.do_include_vectorized({'a': np.array([1, 2])})
    np.testing.assert_array_equal(mask, [False, True])


def test_in_reduce_all_any():
    p_all = in_reduce([in_set({1, 2}, 'a'), in_set({2, 3}, 'a')], all)
    assert p_all.do_include({'a': 2}) and not p_all.do_include({'a': 1})
    p_any = in_reduce([in_set({1}, 'a'), in_set({3}, 'a')], any)
    mask = p_any.do_include_vectorized({'a': np.array([1, 2, 3])})
    np.testing.assert_array_equal(mask, [True, False, True])


def test_in_pseudorandom_split_partitions_and_determinism():
    splits = [0.5, 0.3, 0.2]
    preds = [in_pseudorandom_split(splits, i, 'id') for i in range(3)]
    ids = ['row-{}'.format(i) for i in range(2000)]
    assigned = []
    for v in ids:
        hits = [i for i, p in enumerate(preds) if p.do_include({'id': v})]
        assert len(hits) == 1  # exactly one partition
        assigned.append(hits[0])
    # deterministic
    again = [next(i for i, p in enumerate(preds) if p.do_include({'id': v}))
             for v in ids]
    assert assigned == again
    # fractions are approximately honored
    frac0 = assigned.count(0) / len(assigned)
    assert 0.4 < frac0 < 0.6


def test_pseudorandom_split_vectorized_matches_scalar():
    import numpy as np
    from petastorm_amd.predicates import in_pseudorandom_split
    p = in_pseudorandom_split([0.6, 0.4], 0, 'id')
    ids = np.arange(500)
    vec = p.do_include_vectorized({'id': ids})
    scalar = np.array([p.do_include({'id': i}) for i in ids])
    np.testing.assert_array_equal(vec, scalar)
    assert 200 < vec.sum() < 400  # roughly the 60% fraction


def test_pseudorandom_split_upstream_bucket_contract():
    """Split membership must equal upstream petastorm's rule:
    int(md5(str(v)).hexdigest(), 16) % sys.maxsize vs fraction*(maxsize-1)
    boundaries (reference predicates.py:39-41,172-182)."""
    import hashlib
    import sys
    splits = [0.5, 0.2, 0.3]
    borders = [sum(splits[:i + 1]) for i in range(len(splits))]
    ids = ['vol-%d' % i for i in range(500)] + [str(i) for i in range(500)]
    for idx in range(3):
        p = in_pseudorandom_split(splits, idx, 'id')
        lo = (borders[idx - 1] if idx else 0) * (sys.maxsize - 1)
        hi = borders[idx] * (sys.maxsize - 1)
        for v in ids:
            bucket = int(hashlib.md5(str(v).encode('utf-8')).hexdigest(),
                         16) % sys.maxsize
            assert p.do_include({'id': v}) == (lo <= bucket < hi)


def test_pseudorandom_split_missing_field_raises():
    p = in_pseudorandom_split([0.5, 0.5], 0, 'id')
    with pytest.raises(ValueError):
        p.do_include({'other': 1})
