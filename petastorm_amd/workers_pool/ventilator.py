"""Ventilator: the epoch engine that feeds work items to a pool with
backpressure.

Parity: /root/reference/petastorm/workers_pool/ventilator.py:26-174.

* ``Ventilator`` ABC (:26-52)
* ``ConcurrentVentilator``: ventilates the item list ``iterations`` times
  (``None`` = infinite) from a daemon thread, optional per-epoch permutation
  (seeded rng when ``random_seed`` given), backpressure via
  ``max_ventilation_queue_size`` (:64-168)
* ``reset()`` re-arms a completed ventilator for another full run (:128-137)
"""

import random
import threading


class Ventilator(object):
    def __init__(self, ventilate_fn):
        self._ventilate_fn = ventilate_fn

    def start(self):
        raise NotImplementedError()

    def processed_item(self):
        raise NotImplementedError()

    def completed(self):
        raise NotImplementedError()

    def stop(self):
        raise NotImplementedError()


class ConcurrentVentilator(Ventilator):
    def __init__(self, ventilate_fn, items, iterations=1,
                 randomize_item_order=False, random_seed=None,
                 max_ventilation_queue_size=None,
                 ventilation_interval=0.01):
        super(ConcurrentVentilator, self).__init__(ventilate_fn)
        if iterations is not None and (not isinstance(iterations, int) or iterations < 1):
            raise ValueError('iterations must be a positive integer or None; got {!r}'
                             .format(iterations))
        self._items = list(items)
        self._iterations_orig = iterations
        self._iterations_remaining = iterations
        self._randomize_item_order = randomize_item_order
        self._random_seed = random_seed
        self._max_queue = max_ventilation_queue_size or len(self._items) or 1
        self._interval = ventilation_interval

        self._in_flight = 0
        self._in_flight_cv = threading.Condition()
        self._stop_event = threading.Event()
        self._completed = threading.Event()
        self._thread = None
        self._epoch = 0
        # deferred-start support (reader checkpoint fast-forward): while
        # held, start() only records the request; release() performs it
        self._hold = False
        self._start_requested = False
        self._skip_once = 0

    # ------------------------------------------------------------------
    def start(self):
        if self._hold:
            self._start_requested = True
            return
        if self._thread is not None and self._thread.is_alive():
            raise RuntimeError('Ventilator is already running')
        self._stop_event.clear()
        self._completed.clear()
        self._thread = threading.Thread(target=self._ventilate, daemon=True,
                                        name='petastorm-amd-ventilator')
        self._thread.start()

    def hold(self):
        """Defer the next start() until release() — lets a reader position
        the ventilator (fast_forward) before any item flows."""
        self._hold = True

    def release(self):
        self._hold = False
        if self._start_requested:
            self._start_requested = False
            self.start()

    def fast_forward(self, epoch, skip_items):
        """Position BEFORE any ventilation: begin at ``epoch`` (the
        per-epoch permutation uses random_seed+epoch, and the remaining
        iteration budget shrinks accordingly) and skip the first
        ``skip_items`` items of that epoch.  Only valid while held /
        not yet running."""
        if self._thread is not None and self._thread.is_alive():
            raise RuntimeError('fast_forward requires a not-yet-running '
                               'ventilator')
        self._epoch = epoch
        if self._iterations_remaining is not None:
            self._iterations_remaining = max(
                0, self._iterations_orig - epoch)
        self._skip_once = int(skip_items)

    def _ventilate(self):
        while not self._stop_event.is_set():
            if self._iterations_remaining is not None and self._iterations_remaining <= 0:
                break
            items = list(self._items)
            if self._randomize_item_order:
                rng = random.Random(
                    None if self._random_seed is None
                    else self._random_seed + self._epoch)
                rng.shuffle(items)
            if self._skip_once:
                items = items[self._skip_once:]
                self._skip_once = 0
            for item in items:
                # backpressure (reference ventilator.py:155-157)
                with self._in_flight_cv:
                    while self._in_flight >= self._max_queue and not self._stop_event.is_set():
                        self._in_flight_cv.wait(timeout=self._interval)
                    if self._stop_event.is_set():
                        break
                    self._in_flight += 1
                self._ventilate_fn(**item) if isinstance(item, dict) \
                    else self._ventilate_fn(item)
            self._epoch += 1
            if self._iterations_remaining is not None:
                self._iterations_remaining -= 1
        self._completed.set()

    # ------------------------------------------------------------------
    def processed_item(self):
        with self._in_flight_cv:
            self._in_flight = max(0, self._in_flight - 1)
            self._in_flight_cv.notify_all()

    def completed(self):
        """True when no further items will ever be ventilated."""
        return self._completed.is_set() and \
            (self._thread is None or not self._thread.is_alive())

    def reset(self):
        """Re-arm for another full ``iterations`` run
        (reference ventilator.py:128-137)."""
        if self._thread is not None and self._thread.is_alive():
            raise RuntimeError('Can not reset a ventilator that is still running')
        self._iterations_remaining = self._iterations_orig
        with self._in_flight_cv:
            self._in_flight = 0
        self.start()

    def stop(self):
        self._stop_event.set()
        with self._in_flight_cv:
            self._in_flight_cv.notify_all()
        if self._thread is not None:
            self._thread.join(timeout=10)
