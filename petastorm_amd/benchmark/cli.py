"""``petastorm-amd-throughput`` CLI.

Parity: /root/reference/petastorm/benchmark/cli.py (defaults at :42-78: 3
workers, 200 warmup, 1000 measured cycles, q=500, min-after-dequeue=0.8q).
"""

import argparse
import logging
import sys

from petastorm_amd.benchmark.throughput import (WorkerPoolType,
                                                reader_throughput)


def main(args=None):
    ap = argparse.ArgumentParser(
        description='petastorm_amd reader throughput benchmark')
    ap.add_argument('dataset_url', help='e.g. file:///tmp/hello_world')
    ap.add_argument('--field-regex', nargs='+', default=None,
                    help='read only fields matching these regexes')
    ap.add_argument('-w', '--warmup-cycles', type=int, default=200)
    ap.add_argument('-m', '--measure-cycles', type=int, default=1000)
    ap.add_argument('-p', '--pool-type',
                    choices=[WorkerPoolType.THREAD, WorkerPoolType.PROCESS,
                             WorkerPoolType.NONE],
                    default=WorkerPoolType.THREAD)
    ap.add_argument('-l', '--loaders-count', type=int, default=3)
    ap.add_argument('-r', '--read-method', choices=['python', 'batch'],
                    default='python')
    ap.add_argument('-q', '--shuffling-queue-size', type=int, default=500)
    ap.add_argument('-d', '--min-after-dequeue', type=int, default=400)
    ap.add_argument('--device', default=None,
                    help="'cuda' for the MI355X GPU decode pipeline")
    ap.add_argument('--spawn-new-process', action='store_true',
                    help='measure in a fresh process for clean RSS '
                         '(reference throughput.py:144-149)')
    ap.add_argument('-v', '--verbose', action='store_true')
    args = ap.parse_args(args)
    if args.verbose:
        logging.basicConfig(level=logging.DEBUG)

    result = reader_throughput(
        args.dataset_url, args.field_regex,
        warmup_cycles_count=args.warmup_cycles,
        measure_cycles_count=args.measure_cycles,
        pool_type=args.pool_type, loaders_count=args.loaders_count,
        read_method=args.read_method, device=args.device,
        shuffling_queue_size=args.shuffling_queue_size,
        min_after_dequeue=args.min_after_dequeue,
        spawn_new_process=args.spawn_new_process)
    rss = result.memory_info.rss / 2 ** 20 if result.memory_info else 0.0
    print('Throughput: {:.2f} samples/sec; RAM {:.2f} MB (rss); '
          'CPU {:.1f}%'.format(result.samples_per_second, rss,
                               result.cpu or 0.0))
    return 0


if __name__ == '__main__':
    sys.exit(main())
