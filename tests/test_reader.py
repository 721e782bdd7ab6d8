

def test_cpu_reader_state_dict_resume(tmp_path, scalar_dataset):
    """Deterministic fast-forward resume on the CPU pool path (the GPU
    reader has an exact-cursor state_dict; reference has neither)."""
    from petastorm_amd import make_reader
    url = scalar_dataset['url']
    kwargs = dict(reader_pool_type='thread', workers_count=3,
                  shuffle_row_groups=True, seed=11, num_epochs=2)
    with make_reader(url, **kwargs) as r1:
        first = [int(next(r1).id) for _ in range(120)]
        state = r1.state_dict()
        rest1 = [int(row.id) for row in r1]
    with make_reader(url, **kwargs) as r2:
        r2.load_state_dict(state)
        rest2 = [int(row.id) for row in r2]
    assert state['rows_consumed'] == 120
    assert rest1 == rest2
    assert len(first) + len(rest1) == 2 * 500


def test_cpu_reader_state_dict_requires_determinism(scalar_dataset):
    from petastorm_amd import make_reader
    import pytest as _pytest
    with make_reader(scalar_dataset['url'], shuffle_row_groups=True,
                     seed=None) as r:
        with _pytest.raises(NotImplementedError):
            r.state_dict()
