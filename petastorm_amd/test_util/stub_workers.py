"""Stub workers for pool stress tests (importable from spawned worker
processes — reference keeps its equivalents in
workers_pool/tests/stub_workers.py)."""
import os

from petastorm_amd.workers_pool.worker_base import WorkerBase


class EchoWorker(WorkerBase):
    """Publishes its input back."""

    def process(self, item):
        self.publish_func(('echo', self.worker_id, item))


class FailingWorker(WorkerBase):
    """Raises on items divisible by the configured modulus."""

    def process(self, item):
        if item % (self.args or 3) == 0:
            raise ValueError('boom on %d' % item)
        self.publish_func(item)


class CrashingWorker(WorkerBase):
    """Hard-kills its own process on a trigger item (models a segfault /
    OOM kill — no exception message ever reaches the results queue)."""

    def process(self, item):
        if item == 'die':
            os._exit(17)
        self.publish_func(item)


class SlowWorker(WorkerBase):
    def process(self, item):
        import time
        time.sleep(0.2)
        self.publish_func(item)


class DiesOnInitWorker(WorkerBase):
    """Exits the process during construction (spawn-bootstrap failure
    stand-in for the startup fail-fast test)."""

    def __init__(self, worker_id, publish_func, args):
        import os
        os._exit(3)
