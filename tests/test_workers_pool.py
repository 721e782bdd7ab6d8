"""Pool + ventilator mechanisms (parity: reference
workers_pool/tests/test_workers_pool.py, test_ventilator.py)."""
import threading
import time

import pytest

from petastorm_amd.workers_pool import EmptyResultError
from petastorm_amd.workers_pool.dummy_pool import DummyPool
from petastorm_amd.workers_pool.thread_pool import ThreadPool
from petastorm_amd.workers_pool.ventilator import ConcurrentVentilator
from petastorm_amd.workers_pool.worker_base import WorkerBase


class EchoWorker(WorkerBase):
    def process(self, value):
        self.publish_func(value * 10)


class FailingWorker(WorkerBase):
    def process(self, value):
        if value == 3:
            raise RuntimeError('boom on 3')
        self.publish_func(value)


class SlowWorker(WorkerBase):
    def process(self, value):
        time.sleep(0.01)
        self.publish_func(value)


@pytest.mark.parametrize('pool_factory', [lambda: ThreadPool(3), DummyPool])
def test_pool_roundtrip(pool_factory):
    pool = pool_factory()
    pool.start(EchoWorker)
    for i in range(20):
        pool.ventilate(i)
    results = sorted(pool.get_results() for _ in range(20))
    assert results == [i * 10 for i in range(20)]
    pool.stop()
    pool.join()


def test_thread_pool_deterministic_order():
    """Round-robin readout keeps ventilation order (reference
    thread_pool.py:172-218)."""
    for _ in range(3):
        pool = ThreadPool(4)
        pool.start(SlowWorker)
        for i in range(16):
            pool.ventilate(i)
        got = [pool.get_results() for _ in range(16)]
        assert got == list(range(16))
        pool.stop()
        pool.join()


def test_empty_result_error_with_ventilator():
    pool = ThreadPool(2)
    vent = ConcurrentVentilator(ventilate_fn=None, items=[{'value': i} for i in range(7)],
                                iterations=2)
    vent._ventilate_fn = pool.ventilate
    pool.start(EchoWorker, ventilator=vent)
    got = []
    with pytest.raises(EmptyResultError):
        while True:
            got.append(pool.get_results())
    assert sorted(got) == sorted([i * 10 for i in range(7)] * 2)
    pool.stop()
    pool.join()


def test_worker_exception_reraised():
    pool = ThreadPool(2)
    pool.start(FailingWorker)
    for i in range(5):
        pool.ventilate(i)
    with pytest.raises(RuntimeError, match='boom on 3'):
        for _ in range(5):
            pool.get_results()


def test_ventilator_epochs_and_backpressure():
    ventilated = []
    lock = threading.Lock()

    def fn(value):
        with lock:
            ventilated.append(value)

    vent = ConcurrentVentilator(fn, items=[{'value': i} for i in range(4)],
                                iterations=3, max_ventilation_queue_size=2)
    vent.start()
    # backpressure: without processed_item calls, at most 2 in flight
    time.sleep(0.2)
    with lock:
        assert len(ventilated) == 2
    for _ in range(12):
        vent.processed_item()
        time.sleep(0.01)
    deadline = time.time() + 5
    while not vent.completed() and time.time() < deadline:
        time.sleep(0.01)
    assert vent.completed()
    assert len(ventilated) == 12  # 4 items x 3 epochs


def test_ventilator_seeded_permutation_deterministic():
    def collect(seed):
        out = []
        vent = ConcurrentVentilator(lambda value: out.append(value),
                                    items=[{'value': i} for i in range(10)],
                                    iterations=2, randomize_item_order=True,
                                    random_seed=seed,
                                    max_ventilation_queue_size=100)
        vent.start()
        deadline = time.time() + 5
        while not vent.completed() and time.time() < deadline:
            time.sleep(0.01)
        return out

    assert collect(1) == collect(1)
    assert collect(1) != collect(2)


def test_ventilator_reset():
    out = []
    vent = ConcurrentVentilator(lambda value: out.append(value),
                                items=[{'value': i} for i in range(5)],
                                iterations=1, max_ventilation_queue_size=100)
    vent.start()
    time.sleep(0.3)
    assert vent.completed() and len(out) == 5
    vent.reset()
    deadline = time.time() + 5
    while not vent.completed() and time.time() < deadline:
        time.sleep(0.01)
    assert len(out) == 10


def test_ventilator_invalid_iterations():
    with pytest.raises(ValueError):
        ConcurrentVentilator(lambda v: None, [], iterations=0)


def test_reader_diagnostics_contract(tmp_path):
    """reader.diagnostics exposes the pool counters the benchmark harness
    reads (reference thread_pool.py:258-263, process_pool.py:303-312)."""
    from petastorm_amd import make_reader
    from petastorm_amd.test_util.dataset_gen import create_scalar_dataset
    url = 'file://' + str(tmp_path / 'diag')
    create_scalar_dataset(url, num_rows=100, rowgroup_size=50)
    for pool in ('dummy', 'thread'):
        with make_reader(url, reader_pool_type=pool, num_epochs=1) as r:
            list(r)
            d = r.diagnostics
            assert d['items_ventilated'] == 2
            assert d['items_processed'] == 2
