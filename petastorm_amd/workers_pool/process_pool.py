"""ProcessPool: spawned worker processes for decode parallelism that must
escape the GIL on the CPU path.

Parity: /root/reference/petastorm/workers_pool/process_pool.py (424 LoC of
ZeroMQ PUSH/PUB/PULL plumbing).  Re-designed, not translated: this
environment has no libzmq, and the MI355X framework's hot path is the
HIP-stream GPU pipeline (petastorm_amd/gpu/), so the process pool's only job
is CPU-side decode scale-out.  ``multiprocessing`` with the ``spawn`` start
method (reference also spawns rather than forks, process_pool.py:330-413)
and two shared queues replace the three zmq sockets:

* ventilation: ``mp.Queue`` (reference PUSH socket, :179-185)
* results:     ``mp.Queue`` carrying (serialized payload | control msg)
  (reference PULL socket + multipart messages, :251-270)
* startup handshake: each worker posts a started marker; ``start()`` waits
  with a timeout (reference _WORKER_STARTED_INDICATOR, :207-213)
* orphan protection: a monitor thread in each worker kills the process when
  the parent pid dies (reference _monitor_thread_function, :320-327)

Ordering note: like the reference's process pool, result order is arrival
order (not deterministic round-robin) — use ThreadPool or DummyPool when
byte-exact deterministic ordering is required.
"""

import multiprocessing as mp
import os
import threading
import time
import traceback

from petastorm_amd.workers_pool import (EmptyResultError,
                                        TimeoutWaitingForResultError)
from petastorm_amd.reader_impl.serializers import PickleSerializer

_STOP = '__petastorm_amd_stop__'
_STARTED = '__petastorm_amd_worker_started__'
# Spawned workers each import the full torch stack; on a contended host a
# wide pool can need minutes.  A dead worker fails the handshake
# immediately, so a generous cap only delays reporting genuine hangs.
_WORKER_START_TIMEOUT_S = int(os.environ.get(
    'PSA_WORKER_START_TIMEOUT_S', '180'))


def _orphan_monitor(parent_pid):
    """Kill this worker if the parent dies (reference :320-327)."""
    while True:
        if os.getppid() != parent_pid:
            os._exit(1)
        time.sleep(1.0)


def _worker_main(worker_id, worker_class, worker_args, work_q, results_q,
                 serializer, parent_pid):
    threading.Thread(target=_orphan_monitor, args=(parent_pid,),
                     daemon=True).start()

    def publish(payload):
        results_q.put(('payload', serializer.serialize(payload)))

    worker = worker_class(worker_id, publish, worker_args)
    results_q.put(('control', _STARTED))
    while True:
        item = work_q.get()
        if item == _STOP:
            break
        try:
            if isinstance(item, dict):
                worker.process(**item)
            elif isinstance(item, tuple):
                worker.process(*item)
            else:
                worker.process(item)
            results_q.put(('control', 'processed'))
        except Exception as e:  # noqa: BLE001 - forwarded to consumer
            results_q.put(('exception', (e, traceback.format_exc())))
            results_q.put(('control', 'processed'))
    worker.shutdown()


class ProcessPool(object):
    def __init__(self, workers, serializer=None):
        self.workers_count = workers
        self._serializer = serializer or PickleSerializer()
        self._ctx = mp.get_context('spawn')
        self._work_q = None
        self._results_q = None
        self._procs = []
        self._ventilator = None
        self._ventilated = 0
        self._processed = 0
        self._count_lock = threading.Lock()
        self._stopped = False

    def start(self, worker_class, worker_args=None, ventilator=None):
        self._work_q = self._ctx.Queue()
        self._results_q = self._ctx.Queue()
        parent_pid = os.getpid()
        for i in range(self.workers_count):
            p = self._ctx.Process(
                target=_worker_main,
                args=(i, worker_class, worker_args, self._work_q,
                      self._results_q, self._serializer, parent_pid),
                daemon=True)
            p.start()
            self._procs.append(p)
        # startup handshake (reference :207-213)
        started = 0
        deadline = time.time() + _WORKER_START_TIMEOUT_S
        pending = []
        while started < self.workers_count:
            dead = [p for p in self._procs if p.exitcode is not None]
            if dead:
                raise RuntimeError(
                    'Worker process(es) died during startup (exitcodes '
                    '{})'.format([p.exitcode for p in dead]))
            remaining = deadline - time.time()
            if remaining <= 0:
                raise TimeoutWaitingForResultError(
                    'Timed out waiting for {} worker processes to start '
                    '(raise PSA_WORKER_START_TIMEOUT_S for contended '
                    'hosts)'.format(self.workers_count - started))
            try:
                kind, msg = self._results_q.get(timeout=min(remaining, 0.5))
            except Exception:
                continue
            if kind == 'control' and msg == _STARTED:
                started += 1
            else:
                pending.append((kind, msg))
        self._pending = pending
        self._ventilator = ventilator
        if ventilator is not None:
            ventilator.start()

    def ventilate(self, *args, **kwargs):
        item = kwargs if kwargs else (args if len(args) != 1 else args[0])
        with self._count_lock:
            self._ventilated += 1
        self._work_q.put(item)

    def get_results(self):
        while True:
            if self._stopped:
                raise EmptyResultError('Pool was stopped')
            if self._pending:
                kind, msg = self._pending.pop(0)
            else:
                if self._all_done():
                    raise EmptyResultError('No more work')
                try:
                    kind, msg = self._results_q.get(timeout=0.05)
                except Exception:
                    # a crashed worker (segfault, OOM-kill, os._exit) never
                    # posts its 'processed' control message — detect the
                    # death instead of spinning forever (reference keeps
                    # workers under a zmq socket whose closure surfaces
                    # similarly, process_pool.py:289-294)
                    dead = [p for p in self._procs
                            if p.exitcode is not None and p.exitcode != 0]
                    if dead and self._results_q.empty():
                        self.stop()
                        self.join()
                        raise RuntimeError(
                            'Worker process(es) died with exit code(s) {} '
                            'before finishing their work items'
                            .format([p.exitcode for p in dead]))
                    continue
            if kind == 'payload':
                return self._serializer.deserialize(msg)
            if kind == 'exception':
                exc, tb = msg
                self.stop()
                self.join()
                raise exc
            if kind == 'control' and msg == 'processed':
                with self._count_lock:
                    self._processed += 1
                if self._ventilator is not None:
                    self._ventilator.processed_item()

    def _all_done(self):
        with self._count_lock:
            if self._processed < self._ventilated:
                return False
        if self._ventilator is not None and not self._ventilator.completed():
            return False
        return self._results_q.empty()

    def stop(self):
        if self._stopped:
            return
        self._stopped = True
        if self._ventilator is not None:
            self._ventilator.stop()
        for _ in self._procs:
            try:
                self._work_q.put(_STOP)
            except Exception:
                pass

    def join(self):
        for p in self._procs:
            p.join(timeout=15)
            if p.is_alive():
                p.terminate()
                p.join(timeout=5)

    @property
    def diagnostics(self):
        return {
            'items_ventilated': self._ventilated,
            'items_processed': self._processed,
        }
