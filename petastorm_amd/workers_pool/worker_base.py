"""Worker protocol.

Parity: /root/reference/petastorm/workers_pool/worker_base.py:18-35.
"""


class WorkerBase(object):
    def __init__(self, worker_id, publish_func, args):
        """
        :param worker_id: ordinal of this worker in the pool
        :param publish_func: callable(payload) delivering a result to the pool
        :param args: pool-wide worker args tuple
        """
        self.worker_id = worker_id
        self.publish_func = publish_func
        self.args = args

    def process(self, *args, **kwargs):
        raise NotImplementedError()

    def shutdown(self):
        """Called once when the pool stops (optional cleanup hook)."""
