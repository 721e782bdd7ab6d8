"""DummyPool: single-threaded, in-caller-thread execution for debugging,
profiling and fully deterministic tests.

Parity: /root/reference/petastorm/workers_pool/dummy_pool.py:20-91 —
``process()`` runs lazily inside ``get_results`` (:50-80) so ordinary
profilers see the worker code.
"""

from collections import deque

from petastorm_amd.workers_pool import (EmptyResultError,
                                        VentilatedItemProcessedMessage)


class DummyPool(object):
    def __init__(self, *_args, **_kwargs):
        self.workers_count = 1
        self._worker = None
        self._ventilator = None
        self._items = deque()
        self._results = deque()
        self._stopped = False
        self._ventilated = 0
        self._processed = 0

    def start(self, worker_class, worker_args=None, ventilator=None):
        self._worker = worker_class(0, self._results.append, worker_args)
        self._ventilator = ventilator
        if ventilator is not None:
            ventilator.start()

    def ventilate(self, *args, **kwargs):
        item = kwargs if kwargs else (args if len(args) != 1 else args[0])
        self._ventilated += 1
        self._items.append(item)

    def get_results(self):
        while True:
            while self._results:
                msg = self._results.popleft()
                if isinstance(msg, VentilatedItemProcessedMessage):
                    continue
                return msg
            if self._items:
                item = self._items.popleft()
                try:
                    if isinstance(item, dict):
                        self._worker.process(**item)
                    elif isinstance(item, tuple):
                        self._worker.process(*item)
                    else:
                        self._worker.process(item)
                finally:
                    self._processed += 1
                    if self._ventilator is not None:
                        self._ventilator.processed_item()
                continue
            if self._ventilator is not None and not self._ventilator.completed():
                # let the ventilator thread push more items
                import time
                time.sleep(0.001)
                continue
            raise EmptyResultError('No more work')

    def stop(self):
        self._stopped = True
        if self._ventilator is not None:
            self._ventilator.stop()

    def join(self):
        if self._worker is not None:
            self._worker.shutdown()

    @property
    def diagnostics(self):
        return {'output_queue_size': len(self._results),
                'items_ventilated': self._ventilated,
                'items_processed': self._processed}
