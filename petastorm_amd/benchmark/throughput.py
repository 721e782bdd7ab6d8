"""Reader throughput benchmark harness.

Parity: /root/reference/petastorm/benchmark/throughput.py.

* ``BenchmarkResult(time_mean, samples_per_second, memory_info, cpu)``
  (reference :38)
* warmup + measured ``next(reader)`` cycles (reference :68-90)
* RSS / CPU%% via psutil (reference :76-88)

The reference re-spawns itself in a fresh process for clean RSS numbers
(reference :144-149); here ``spawn_new_process=True`` uses multiprocessing
spawn for the same effect.
"""

import time
from collections import namedtuple

BenchmarkResult = namedtuple('BenchmarkResult',
                             ['time_mean', 'samples_per_second',
                              'memory_info', 'cpu'])

#: Reference-name parity (throughput.py): read_method values.
ReadMethod = namedtuple('ReadMethods', ['PYTHON', 'TF'])('python', 'tf')

WorkerPoolType = namedtuple('WorkerPoolTypes', ['THREAD', 'PROCESS', 'NONE'])(
    'thread', 'process', 'dummy')


def _time_warmup_and_work(reader, warmup_cycles, measure_cycles):
    """reference throughput.py:68-90"""
    for _ in range(warmup_cycles):
        next(reader)
    t0 = time.perf_counter()
    count = 0
    for _ in range(measure_cycles):
        row = next(reader)
        if getattr(reader, 'batched_output', False):
            first = row[0]
            count += len(first)
        else:
            count += 1
    elapsed = time.perf_counter() - t0

    try:
        import psutil
        proc = psutil.Process()
        memory_info = proc.memory_info()
        cpu = proc.cpu_percent()
    except ImportError:  # pragma: no cover
        memory_info, cpu = None, None
    return BenchmarkResult(time_mean=elapsed / measure_cycles,
                           samples_per_second=count / elapsed,
                           memory_info=memory_info, cpu=cpu)


def reader_throughput(dataset_url, field_regex=None, warmup_cycles_count=200,
                      measure_cycles_count=1000,
                      pool_type=WorkerPoolType.THREAD, loaders_count=3,
                      read_method='python',
                      shuffling_queue_size=500, min_after_dequeue=400,
                      device=None, spawn_new_process=False):
    """Benchmark ``make_reader`` (or the GPU batch reader) on a dataset.

    reference throughput.py:112-172.  ``read_method`` 'python' iterates rows
    (reference default); 'batch' uses make_batch_reader.
    """
    if spawn_new_process:
        # clean-memory measurement in a fresh interpreter (reference :144-149)
        import multiprocessing as mp
        ctx = mp.get_context('spawn')
        with ctx.Pool(1) as pool:
            return pool.apply(reader_throughput, (dataset_url,), dict(
                field_regex=field_regex,
                warmup_cycles_count=warmup_cycles_count,
                measure_cycles_count=measure_cycles_count,
                pool_type=pool_type, loaders_count=loaders_count,
                read_method=read_method,
                shuffling_queue_size=shuffling_queue_size,
                min_after_dequeue=min_after_dequeue, device=device,
                spawn_new_process=False))

    from petastorm_amd import make_batch_reader, make_reader

    if read_method == 'batch' or device is not None:
        reader = make_batch_reader(dataset_url, schema_fields=field_regex,
                                   reader_pool_type=pool_type,
                                   workers_count=loaders_count,
                                   num_epochs=None, device=device)
    elif read_method == 'python':
        reader = make_reader(dataset_url, schema_fields=field_regex,
                             reader_pool_type=pool_type,
                             workers_count=loaders_count, num_epochs=None)
    else:
        raise ValueError('Unknown read_method {!r}'.format(read_method))
    try:
        return _time_warmup_and_work(reader, warmup_cycles_count,
                                     measure_cycles_count)
    finally:
        reader.stop()
        reader.join()
