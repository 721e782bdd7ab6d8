"""Worker-pool protocol types.

Parity: /root/reference/petastorm/workers_pool/__init__.py:16-26.
"""


class EmptyResultError(Exception):
    """Raised by ``pool.get_results()`` when all ventilated work is done and
    no more results will arrive."""


class TimeoutWaitingForResultError(Exception):
    """Raised when a bounded wait for a result expires."""


class VentilatedItemProcessedMessage(object):
    """Control message a worker emits after fully processing one ventilated
    item (reference workers_pool/__init__.py:24-26)."""


class WorkerExceptionMessage(object):
    """Carries a worker-side exception (+ formatted traceback) to the
    consumer, where it is re-raised (reference thread_pool.py:67-72,
    211-214)."""

    def __init__(self, exception, traceback_str):
        self.exception = exception
        self.traceback_str = traceback_str
