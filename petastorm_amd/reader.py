"""Reader API and orchestration: ``make_reader`` / ``make_batch_reader`` /
``Reader``.

Parity: /root/reference/petastorm/reader.py.

* ``make_reader`` (reference :60-206): petastorm datasets, row output.
* ``make_batch_reader`` (reference :209-352): columnar batch output; also
  works on plain (non-petastorm) Parquet stores via schema inference.
* ``Reader`` (reference :361-501): opens the dataset, loads/infers the
  schema, filters row groups (predicate / selector / shard modulo), builds a
  ConcurrentVentilator of ``{piece_index, worker_predicate,
  shuffle_row_drop_partition}`` items and starts a worker pool; iterates
  results; ``reset`` (:503-527) / ``stop`` / ``join``.
* Sharding rule ``index % shard_count == cur_shard`` (reference :596) with
  optional seeded pre-shuffle (:589-594), ``NoDataAvailableError`` when a
  shard is empty (:583-585).
* Ventilation backpressure of ``workers x (1 + 3)`` in-flight row-groups
  (reference :45, :489).

MI355X note: ``make_batch_reader(..., device='cuda')`` routes to the GPU
pipeline (petastorm_amd.gpu) where Parquet pages are decoded by HIP kernels
and batches are HBM-resident torch tensors; the pool/ventilator plumbing
below is the CPU path and the planning layer shared by both.
"""

import logging
import warnings

import numpy as np

from petastorm_amd.cache import LocalDiskCache, NullCache
from petastorm_amd.errors import NoDataAvailableError
from petastorm_amd.etl import dataset_metadata as dsm
from petastorm_amd.fs_utils import (get_filesystem_and_path_or_paths,
                                    normalize_dataset_url_or_urls)
from petastorm_amd.ngram import NGram
from petastorm_amd.reader_impl.serializers import PickleSerializer
from petastorm_amd.transform import transform_schema
from petastorm_amd.unischema import match_unischema_fields
from petastorm_amd.workers.batch_worker import (BatchReaderWorker,
                                                BatchWorkerArgs)
from petastorm_amd.workers.row_worker import RowReaderWorker, RowWorkerArgs
from petastorm_amd.workers_pool import EmptyResultError
from petastorm_amd.workers_pool.dummy_pool import DummyPool
from petastorm_amd.workers_pool.process_pool import ProcessPool
from petastorm_amd.workers_pool.thread_pool import ThreadPool
from petastorm_amd.workers_pool.ventilator import ConcurrentVentilator

logger = logging.getLogger(__name__)

#: extra row-groups ventilated beyond one per worker (reference reader.py:45)
_VENTILATE_EXTRA_ROWGROUPS = 3


def _make_cache(cache_type, cache_location, cache_size_limit,
                cache_row_size_estimate, cache_extra_settings):
    """reference reader.py:150-155"""
    if cache_type in (None, 'null'):
        return NullCache()
    if cache_type == 'local-disk':
        if not cache_location or not cache_size_limit:
            raise ValueError('local-disk cache requires cache_location and '
                             'cache_size_limit')
        return LocalDiskCache(cache_location, cache_size_limit,
                              cache_row_size_estimate or 0,
                              **(cache_extra_settings or {}))
    raise ValueError('Unknown cache_type: {!r}'.format(cache_type))


def _make_pool(reader_pool_type, workers_count, results_queue_size,
               serializer=None):
    """reference reader.py:164-175"""
    if reader_pool_type == 'thread':
        return ThreadPool(workers_count, results_queue_size)
    if reader_pool_type == 'process':
        return ProcessPool(workers_count, serializer or PickleSerializer())
    if reader_pool_type == 'dummy':
        return DummyPool()
    raise ValueError('Unknown reader_pool_type: {!r}'.format(reader_pool_type))


def _upstream_compat(seed, shard_seed, ignored):
    """Accept upstream-petastorm kwargs that have no effect here: map
    ``shard_seed`` onto the unified ``seed`` (reference deprecates it the
    same way, reader.py:475-479) and warn about transport/driver knobs this
    framework replaced (zmq/libhdfs3/pyarrow-serialization)."""
    if shard_seed is not None:
        warnings.warn('shard_seed is deprecated; it now sets the unified '
                      'seed (row-group shuffle + sharding + row shuffle)',
                      DeprecationWarning)
        if seed is None:
            seed = shard_seed
    for name, value in ignored.items():
        if value is not None:
            warnings.warn('{} is accepted for upstream-petastorm '
                          'compatibility but has no effect here'
                          .format(name), DeprecationWarning)
    return seed


def _resolve_fs(dataset_url_or_urls, storage_options, filesystem):
    """URL(s) -> (fs, path-or-paths); an explicit ``filesystem`` (upstream
    reader.py:60-77 ``filesystem=``) bypasses scheme resolution."""
    if filesystem is None:
        return get_filesystem_and_path_or_paths(dataset_url_or_urls,
                                                storage_options)
    from urllib.parse import urlparse

    def path_of(u):
        p = urlparse(u)
        return p.path if p.scheme in ('file', '') \
            else (p.netloc + p.path)

    if isinstance(dataset_url_or_urls, list):
        return filesystem, [path_of(u) for u in dataset_url_or_urls]
    return filesystem, path_of(dataset_url_or_urls)


def make_reader(dataset_url,
                schema_fields=None,
                reader_pool_type='thread', workers_count=10,
                results_queue_size=50,
                shuffle_row_groups=True, shuffle_row_drop_partitions=1,
                shuffle_rows=False,
                predicate=None,
                rowgroup_selector=None,
                num_epochs=1,
                cur_shard=None, shard_count=None, seed=None,
                cache_type='null', cache_location=None, cache_size_limit=None,
                cache_row_size_estimate=None, cache_extra_settings=None,
                transform_spec=None, filters=None,
                storage_options=None,
                shard_seed=None, filesystem=None,
                pyarrow_serialize=None, hdfs_driver=None,
                zmq_copy_buffers=None):
    """Row-oriented reader over a petastorm_amd dataset (reference :60-206).

    Each ``next(reader)`` yields one row as a schema-named namedtuple (or a
    ``{timestep: namedtuple}`` dict when ``schema_fields`` is an NGram).
    ``shard_seed``/``pyarrow_serialize``/``hdfs_driver``/``zmq_copy_buffers``
    are accepted for upstream call-site compatibility (deprecation warning);
    ``filesystem`` passes an explicit fsspec/pyarrow filesystem.
    """
    seed = _upstream_compat(seed, shard_seed,
                            dict(pyarrow_serialize=pyarrow_serialize,
                                 hdfs_driver=hdfs_driver,
                                 zmq_copy_buffers=zmq_copy_buffers))
    dataset_url = normalize_dataset_url_or_urls(dataset_url)
    fs, path = _resolve_fs(dataset_url, storage_options, filesystem)
    try:
        dsm.get_schema(fs, path)
    except ValueError:
        warnings.warn('Dataset at {} is missing petastorm_amd metadata; '
                      'consider make_batch_reader for plain Parquet stores'
                      .format(dataset_url))
    cache = _make_cache(cache_type, cache_location, cache_size_limit,
                        cache_row_size_estimate, cache_extra_settings)
    pool = _make_pool(reader_pool_type, workers_count, results_queue_size)
    return Reader(fs, path,
                  worker_class=RowReaderWorker,
                  schema_fields=schema_fields,
                  reader_pool=pool,
                  shuffle_row_groups=shuffle_row_groups,
                  shuffle_row_drop_partitions=shuffle_row_drop_partitions,
                  shuffle_rows=shuffle_rows,
                  predicate=predicate,
                  rowgroup_selector=rowgroup_selector,
                  num_epochs=num_epochs,
                  cur_shard=cur_shard, shard_count=shard_count, seed=seed,
                  cache=cache,
                  transform_spec=transform_spec, filters=filters,
                  batched_output=False)


def make_batch_reader(dataset_url_or_urls,
                      schema_fields=None,
                      reader_pool_type='thread', workers_count=10,
                      results_queue_size=50,
                      shuffle_row_groups=True, shuffle_row_drop_partitions=1,
                      shuffle_rows=False,
                      predicate=None,
                      rowgroup_selector=None,
                      num_epochs=1,
                      cur_shard=None, shard_count=None, seed=None,
                      cache_type='null', cache_location=None,
                      cache_size_limit=None, cache_row_size_estimate=None,
                      cache_extra_settings=None,
                      transform_spec=None, filters=None,
                      decode_codecs=True,
                      storage_options=None,
                      device=None, gpu_options=None,
                      shard_seed=None, filesystem=None,
                      hdfs_driver=None, zmq_copy_buffers=None,
                      convert_early_to_numpy=None):
    """Columnar batch reader (reference :209-352): each ``next(reader)``
    yields one row-group-sized batch as a namedtuple of column arrays.

    Works on plain Parquet stores (schema inferred from the Arrow schema,
    reference :308-316) and on petastorm_amd datasets (codec fields are
    batch-decoded when ``decode_codecs``).

    :param device: ``None``/'cpu' for the worker-pool path, 'cuda' for the
        MI355X HIP pipeline (on-GPU page decode; batches are torch tensors).
    """
    seed = _upstream_compat(seed, shard_seed,
                            dict(hdfs_driver=hdfs_driver,
                                 zmq_copy_buffers=zmq_copy_buffers))
    if convert_early_to_numpy is False:
        warnings.warn('convert_early_to_numpy=False (arrow-table payloads) '
                      'is not supported: this framework always converts in '
                      'the worker (the upstream option True path)',
                      DeprecationWarning)
    dataset_url_or_urls = normalize_dataset_url_or_urls(dataset_url_or_urls)
    fs, path_or_paths = _resolve_fs(dataset_url_or_urls, storage_options,
                                    filesystem)
    if device is not None and str(device).startswith('cuda'):
        from petastorm_amd.gpu.reader import GpuBatchReader
        # loud rejection of options the GPU pipeline does not implement —
        # never silently change semantics the caller asked for
        if shuffle_row_drop_partitions != 1:
            raise NotImplementedError(
                'shuffle_row_drop_partitions is a CPU-worker memory trick '
                '(reference py_dict_reader_worker.py:264-286); the GPU '
                'pipeline decodes whole row-groups in HBM')
        if not decode_codecs:
            raise NotImplementedError(
                'decode_codecs=False (raw encoded values) is a CPU-path '
                'option; the GPU pipeline always decodes')
        gpu_kwargs = dict(gpu_options or {})
        if cache_type not in (None, 'null'):
            if cache_type != 'hbm':
                raise NotImplementedError(
                    "cache_type={!r} on the GPU path; use 'hbm' (decoded "
                    'row-groups cached in HBM3E)'.format(cache_type))
            gpu_kwargs.setdefault('cache_type', 'hbm')
            gpu_kwargs.setdefault('cache_size_limit', cache_size_limit)
        return GpuBatchReader(fs, path_or_paths,
                              schema_fields=schema_fields,
                              shuffle_row_groups=shuffle_row_groups,
                              shuffle_rows=shuffle_rows,
                              predicate=predicate,
                              rowgroup_selector=rowgroup_selector,
                              num_epochs=num_epochs,
                              cur_shard=cur_shard, shard_count=shard_count,
                              seed=seed, transform_spec=transform_spec,
                              filters=filters,
                              device=device, **gpu_kwargs)
    cache = _make_cache(cache_type, cache_location, cache_size_limit,
                        cache_row_size_estimate, cache_extra_settings)
    pool = _make_pool(reader_pool_type, workers_count, results_queue_size)
    return Reader(fs, path_or_paths,
                  worker_class=BatchReaderWorker,
                  schema_fields=schema_fields,
                  reader_pool=pool,
                  shuffle_row_groups=shuffle_row_groups,
                  shuffle_row_drop_partitions=shuffle_row_drop_partitions,
                  shuffle_rows=shuffle_rows,
                  predicate=predicate,
                  rowgroup_selector=rowgroup_selector,
                  num_epochs=num_epochs,
                  cur_shard=cur_shard, shard_count=shard_count, seed=seed,
                  cache=cache,
                  transform_spec=transform_spec, filters=filters,
                  decode_codecs=decode_codecs,
                  batched_output=True)


def _normalize_shuffle_options(shuffle_row_drop_partitions, num_rows_min):
    """reference reader.py:654-664"""
    if not isinstance(shuffle_row_drop_partitions, int) or \
            shuffle_row_drop_partitions < 1:
        raise ValueError('shuffle_row_drop_partitions must be a positive int')
    return shuffle_row_drop_partitions


class Reader(object):
    def __init__(self, filesystem, path_or_paths, worker_class,
                 schema_fields=None, reader_pool=None,
                 shuffle_row_groups=True, shuffle_row_drop_partitions=1,
                 shuffle_rows=False, predicate=None, rowgroup_selector=None,
                 num_epochs=1, cur_shard=None, shard_count=None, seed=None,
                 cache=None, transform_spec=None, filters=None,
                 decode_codecs=True, batched_output=False):
        if (cur_shard is None) != (shard_count is None):
            raise ValueError('cur_shard and shard_count must be used together')
        if num_epochs is not None and (not isinstance(num_epochs, int) or num_epochs < 1):
            raise ValueError('num_epochs must be a positive integer or None')

        self._fs = filesystem
        self._paths = path_or_paths
        self.batched_output = batched_output
        self.last_row_consumed = False
        self._stopped = False

        # --- schema resolution (reference :435-466) ---
        storage_schema, is_petastorm = dsm.infer_or_load_unischema(
            self._fs, self._paths)
        self.is_petastorm_dataset = is_petastorm

        self.ngram = schema_fields if isinstance(schema_fields, NGram) else None
        if self.ngram is not None:
            if batched_output:
                raise NotImplementedError(
                    'NGram is supported by make_reader only (reference '
                    'arrow_reader_worker.py:138-139)')
            self.ngram.resolve_regex_field_names(storage_schema)
            needed = self.ngram.get_field_names_at_all_timesteps()
            view_schema = storage_schema.create_schema_view(
                [storage_schema.fields[n] for n in needed])
        elif schema_fields is not None:
            matched = match_unischema_fields(storage_schema, schema_fields)
            if isinstance(schema_fields, (list, tuple)) and not matched:
                raise ValueError('schema_fields {} matched no fields in the '
                                 'schema'.format(schema_fields))
            view_schema = storage_schema.create_schema_view(matched)
        else:
            view_schema = storage_schema
        self._storage_schema = storage_schema
        self._view_schema = view_schema
        self.transform_spec = transform_spec
        self.schema = transform_schema(view_schema, transform_spec) \
            if transform_spec else view_schema

        # --- row-group planning (reference :471-490) ---
        self._pieces = dsm.load_row_groups(self._fs, self._paths)
        if not self._pieces:
            raise NoDataAvailableError('Dataset has no row groups')
        if filters:
            kept = {p.index for p in dsm.select_pieces_by_filters(
                self._fs, self._pieces, filters)}
        else:
            kept = None
        selected = self._apply_row_group_selector(rowgroup_selector)
        if kept is not None:
            selected = [i for i in selected if i in kept]
        predicate = self._push_down_partition_predicate(predicate, selected)
        if isinstance(predicate, tuple):  # (pushed_down_selection, None)
            selected, predicate = predicate
        selected = self._apply_shard(selected, cur_shard, shard_count, seed,
                                     shuffle_row_groups)

        shuffle_row_drop_partitions = _normalize_shuffle_options(
            shuffle_row_drop_partitions, min(p.num_rows for p in self._pieces))

        # --- pool + ventilator (reference :486-497, :666-682) ---
        self._workers_pool = reader_pool or ThreadPool(10)
        cache = cache or NullCache()
        if worker_class is RowReaderWorker:
            worker_args = RowWorkerArgs(self._fs, storage_schema, view_schema,
                                        self.ngram, self._pieces, cache,
                                        transform_spec, self.schema,
                                        shuffle_rows, seed)
        else:
            worker_args = BatchWorkerArgs(self._fs, storage_schema,
                                          view_schema, self._pieces, cache,
                                          transform_spec, self.schema,
                                          decode_codecs, shuffle_rows, seed)
        items = []
        for piece_index in selected:
            for drop_part in range(shuffle_row_drop_partitions):
                items.append({
                    'piece_index': piece_index,
                    'worker_predicate': predicate,
                    'shuffle_row_drop_partition':
                        (drop_part, shuffle_row_drop_partitions),
                })
        max_q = self._workers_pool.workers_count * (1 + _VENTILATE_EXTRA_ROWGROUPS)
        self._ventilator = ConcurrentVentilator(
            ventilate_fn=None,  # bound below, after pool.start
            items=items,
            iterations=num_epochs,
            randomize_item_order=shuffle_row_groups,
            random_seed=seed,
            max_ventilation_queue_size=max_q)
        self._ventilator._ventilate_fn = self._workers_pool.ventilate
        # hold ventilation until first consumption so load_state_dict can
        # fast-forward the item cursor before any work flows
        self._ventilator.hold()
        self._workers_pool.start(worker_class, worker_args,
                                 ventilator=self._ventilator)
        self._cache = cache
        self._row_buffer = []
        # checkpoint/resume bookkeeping (state_dict); deterministic iteration
        # requires a seed whenever anything shuffles
        self._seed = seed
        self._any_shuffle = bool(shuffle_row_groups or shuffle_rows or
                                 (shuffle_row_drop_partitions or 1) > 1)
        self._rows_consumed = 0
        self._consumption_started = False
        self._num_epochs = num_epochs
        self._shuffle_row_groups_flag = bool(shuffle_row_groups)
        # per-item row counts in VENTILATION ORDER (the O(1) cursor math;
        # exact only when nothing can change per-group row counts: no
        # predicate, no row-drop, no ngram, and no transform func — a
        # TransformSpec func may filter rows, invalidating the metadata
        # counts.  Anything else falls back to consume-and-discard replay.
        self._item_rows = [self._pieces[it['piece_index']].num_rows
                           for it in items]
        self._fast_skip_ok = (predicate is None and
                              (shuffle_row_drop_partitions or 1) == 1 and
                              self.ngram is None and
                              (transform_spec is None or
                               transform_spec.func is None))

    # ------------------------------------------------------------------
    def _push_down_partition_predicate(self, predicate, selected):
        """When every predicate field is a hive-partition key, filter row
        groups by path values and drop the worker-side predicate entirely
        (reference _apply_predicate_to_row_groups, reader.py:620-652)."""
        if predicate is None:
            return None
        try:
            pred_fields = set(predicate.get_fields())
        except Exception:  # noqa: BLE001 - user predicate may defer fields
            return predicate
        if not pred_fields or not all(
                pred_fields <= set(self._pieces[i].partitions)
                for i in selected):
            return predicate
        from petastorm_amd.workers.row_worker import _partition_value

        def keep(i):
            p = self._pieces[i]
            vals = {f: _partition_value(self._storage_schema.fields.get(f),
                                        p.partitions[f])
                    for f in pred_fields}
            return predicate.do_include(vals)

        kept = [i for i in selected if keep(i)]
        if not kept:
            logger.warning('Partition predicate filtered out every row '
                           'group (reference reader.py:567-569)')
        return (kept, None)

    def _apply_row_group_selector(self, selector):
        """reference :599-618"""
        indexes = list(range(len(self._pieces)))
        if selector is None:
            return indexes
        from petastorm_amd.etl.rowgroup_indexing import load_rowgroup_indexes
        index_dict = load_rowgroup_indexes(self._fs, self._paths)
        missing = [n for n in selector.select_index_names()
                   if n not in index_dict]
        if missing:
            raise ValueError('Indexes {} are not available in the dataset '
                             '(available: {})'.format(missing,
                                                      sorted(index_dict)))
        chosen = selector.select_row_groups(index_dict)
        return [i for i in indexes if i in chosen]

    def _apply_shard(self, indexes, cur_shard, shard_count, seed,
                     shuffle_row_groups):
        """reference :573-597"""
        if cur_shard is None:
            return indexes
        if not 0 <= cur_shard < shard_count:
            raise ValueError('cur_shard must be in [0, shard_count)')
        if shard_count > len(indexes):
            raise NoDataAvailableError(
                'Number of row-groups in the dataset ({}) is smaller than '
                'shard_count ({}); at least one shard would see no data'
                .format(len(indexes), shard_count))
        order = list(indexes)
        if shuffle_row_groups and seed is not None:
            rng = np.random.RandomState(seed)
            order = [order[i] for i in rng.permutation(len(order))]
        sharded = [idx for pos, idx in enumerate(order)
                   if pos % shard_count == cur_shard]
        if not sharded:
            logger.warning('Shard %d of %d received no row groups',
                           cur_shard, shard_count)
        return sharded

    # ------------------------------------------------------------------
    def __iter__(self):
        return self

    def __next__(self):
        """reference :708-718"""
        if self._stopped:
            raise StopIteration
        if not self._consumption_started:
            self._consumption_started = True
            self._ventilator.release()
        try:
            if self.batched_output:
                columns = self._workers_pool.get_results()
                self._rows_consumed += 1
                return self.schema.make_namedtuple(**columns)
            while not self._row_buffer:
                self._row_buffer = list(self._workers_pool.get_results())
            row = self._row_buffer.pop(0)
            self._rows_consumed += 1
            if self.ngram is not None:
                return row  # already {timestep: namedtuple}
            return self.schema.make_namedtuple(**row)
        except EmptyResultError:
            self.last_row_consumed = True
            raise StopIteration

    next = __next__

    def reset(self):
        """Restart iteration after exhaustion (reference :503-527)."""
        if not self.last_row_consumed:
            # same open race the reference documents at reader.py:518
            logger.warning('Resetting a reader while mid-iteration may '
                           'produce duplicate or dropped rows')
        self._ventilator.reset()
        self.last_row_consumed = False

    def _check_deterministic(self):
        if self._any_shuffle and self._seed is None:
            raise NotImplementedError(
                'state_dict requires deterministic iteration: pass seed= '
                'or disable shuffling (shuffle_row_groups/shuffle_rows/'
                'shuffle_row_drop_partitions)')

    def state_dict(self):
        """Checkpoint the iterator position (units consumed since
        construction: rows for make_reader, batches for make_batch_reader).

        The reference has no reader checkpointing; the GPU
        reader keeps an exact row-group cursor, while this CPU pool path
        restores by deterministic fast-forward — valid for any seeded (or
        shuffle-free) configuration, with any pool, because the thread
        pool's seeded readout is strict round-robin
        (workers_pool/thread_pool.py) and the dummy pool is inline."""
        self._check_deterministic()
        return {'rows_consumed': self._rows_consumed, 'seed': self._seed,
                'version': 1}

    def load_state_dict(self, state):
        """Fast-forward a FRESH reader (same constructor arguments) to a
        :meth:`state_dict` position.

        Plain configurations (no predicate, no shuffle_row_drop, no
        NGram) restore with an O(1) ITEM CURSOR: whole row groups before
        the checkpoint are never ventilated, decoded or read — only the
        partially-consumed row group (row path) is decoded once to
        discard its leading rows.  Other configurations fall back to
        deterministic consume-and-discard replay."""
        self._check_deterministic()
        if self._rows_consumed or self._consumption_started:
            raise RuntimeError('load_state_dict requires a fresh reader')
        if state.get('seed') != self._seed:
            raise ValueError('state was captured with a different seed')
        target = int(state['rows_consumed'])
        if target and self._fast_skip_ok:
            if self.batched_output:
                # units are row-group batches: one per ventilated item
                per_epoch = len(self._item_rows)
                epoch = target // per_epoch if per_epoch else 0
                skip_items, rem = (target % per_epoch if per_epoch else 0,
                                   0)
            else:
                per_epoch = sum(self._item_rows)
                epoch = target // per_epoch if per_epoch else 0
                rem_rows = target % per_epoch if per_epoch else 0
                order = list(range(len(self._item_rows)))
                if self._shuffle_row_groups_flag:
                    # EXACTLY the ventilator's per-epoch permutation
                    import random as _random
                    _random.Random(
                        None if self._seed is None
                        else self._seed + epoch).shuffle(order)
                skip_items = 0
                for idx in order:
                    n = self._item_rows[idx]
                    if rem_rows < n:
                        break
                    rem_rows -= n
                    skip_items += 1
                rem = rem_rows
            if self._num_epochs is not None and \
                    epoch >= self._num_epochs:
                # checkpoint at/after the end: nothing left to read
                self._ventilator.fast_forward(self._num_epochs, 0)
                self._rows_consumed = target
                self._ventilator.release()
                self._consumption_started = True
                return self
            self._ventilator.fast_forward(epoch, skip_items)
            self._rows_consumed = target - rem
            for _ in range(rem):
                try:
                    next(self)
                except StopIteration:
                    break
            self._consumption_started = True
            self._ventilator.release()
            return self
        for _ in range(target):
            try:
                next(self)
            except StopIteration:
                break
        return self

    def stop(self):
        self._stopped = True
        self._workers_pool.stop()

    def join(self):
        self._workers_pool.join()
        self._cache.cleanup()

    @property
    def diagnostics(self):
        return self._workers_pool.diagnostics

    def __enter__(self):
        return self

    def __exit__(self, exc_type, exc_val, exc_tb):
        self.stop()
        self.join()
