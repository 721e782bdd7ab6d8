"""Process pool end-to-end (spawn is slow; exercised once, not across the
whole e2e matrix)."""
import numpy as np

from petastorm_amd import make_reader


def test_process_pool_roundtrip(test_dataset):
    with make_reader(test_dataset['url'], reader_pool_type='process',
                     workers_count=2, shuffle_row_groups=False) as r:
        rows = list(r)
    assert len(rows) == len(test_dataset['rows'])
    by_id = {int(x.id): x for x in rows}
    src = test_dataset['rows'][0]
    np.testing.assert_array_equal(by_id[int(src['id'])].matrix, src['matrix'])


# ---------------------------------------------------------------------------
# crash / exception / shutdown behavior (reference
# workers_pool/tests/test_workers_pool.py stress suite)
# ---------------------------------------------------------------------------
import pytest

from petastorm_amd.test_util.stub_workers import (CrashingWorker, EchoWorker,
                                                  FailingWorker, SlowWorker)
from petastorm_amd.workers_pool import EmptyResultError
from petastorm_amd.workers_pool.process_pool import ProcessPool


def test_process_pool_many_items_all_arrive():
    pool = ProcessPool(3)
    pool.start(EchoWorker)
    try:
        n = 60
        for i in range(n):
            pool.ventilate(i)
        got = set()
        for _ in range(n):
            tag, wid, item = pool.get_results()
            assert tag == 'echo'
            got.add(item)
        assert got == set(range(n))
        with pytest.raises(EmptyResultError):
            pool.get_results()
        assert pool.diagnostics['items_processed'] == n
    finally:
        pool.stop()
        pool.join()


def test_process_pool_worker_exception_reraised():
    pool = ProcessPool(2)
    pool.start(FailingWorker, worker_args=3)
    for i in [1, 2, 3]:
        pool.ventilate(i)
    with pytest.raises(ValueError, match='boom'):
        for _ in range(3):
            pool.get_results()


def test_process_pool_worker_crash_detected():
    """A hard worker death (os._exit — no exception message possible) must
    surface as a loud error, not an infinite get_results() spin."""
    pool = ProcessPool(1)
    pool.start(CrashingWorker)
    pool.ventilate('die')
    pool.ventilate('never-processed')
    with pytest.raises(RuntimeError, match='died with exit code'):
        for _ in range(2):
            pool.get_results()


def test_process_pool_stop_with_unconsumed_results():
    """Slow-joiner shutdown: stop()+join() must return promptly even with
    results still queued and items in flight (reference :272-301)."""
    import time
    pool = ProcessPool(2)
    pool.start(SlowWorker)
    for i in range(10):
        pool.ventilate(i)
    pool.get_results()  # consume one, leave the rest queued/in-flight
    t0 = time.time()
    pool.stop()
    pool.join()
    assert time.time() - t0 < 20
    with pytest.raises(EmptyResultError):
        pool.get_results()


def test_serializers_roundtrip():
    """PickleSerializer and ArrowTableSerializer round-trip worker payloads
    (reference pickle_serializer.py:17-23, arrow_table_serializer.py:22-33)."""
    import numpy as np
    import pyarrow as pa
    from petastorm_amd.reader_impl.serializers import (ArrowTableSerializer,
                                                       PickleSerializer)
    rows = [{'a': np.arange(4), 'b': 'text'}, {'a': np.zeros(2), 'b': None}]
    ps = PickleSerializer()
    back = ps.deserialize(ps.serialize(rows))
    assert back[1]['b'] is None
    np.testing.assert_array_equal(back[0]['a'], rows[0]['a'])

    table = pa.table({'x': pa.array([1, 2, 3], type=pa.int32()),
                      's': pa.array(['p', 'q', None])})
    ats = ArrowTableSerializer()
    data = ats.serialize(table)
    assert isinstance(data, bytes)
    t2 = ats.deserialize(data)
    assert t2.equals(table)


def test_startup_dead_worker_fails_fast():
    """A worker that dies during construction fails start() immediately
    with its exitcode — not after the full startup timeout."""
    import time
    from petastorm_amd.test_util.stub_workers import DiesOnInitWorker
    from petastorm_amd.workers_pool.process_pool import ProcessPool
    pool = ProcessPool(2)
    t0 = time.monotonic()
    with pytest.raises(RuntimeError, match='died during startup'):
        pool.start(DiesOnInitWorker)
    assert time.monotonic() - t0 < 60  # far below the 180 s timeout
    pool.stop()
    pool.join()
