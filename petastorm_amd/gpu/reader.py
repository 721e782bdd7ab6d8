"""GpuBatchReader: the MI355X HIP pipeline behind
``make_batch_reader(..., device='cuda')``.

The reference's worker pools (thread/process + zmq) become:

* an IO prefetch thread filling *pinned host buffers* with raw row-group
  bytes (depth-2 queue = the ventilator's backpressure, reference
  reader.py:45,489)
* async H2D copies + HIP decode kernels on the current stream
  (petastorm_amd.gpu.decoder) — the decode "workers" are stream slots on the
  same device
* batches are HBM-resident torch tensors; the shuffling queue and batching
  happen on-device in BatchedDataLoader

Sharding follows the reference rule ``index % shard_count == cur_shard``
(reference reader.py:573-597), the per-epoch permutation is broadcast from
rank 0 over RCCL when torch.distributed is initialized, and epoch
boundaries all-gather per-rank row counts (petastorm_amd.parallel.epochs).
"""

import logging
import contextlib
import os
import queue
import threading
import time

import numpy as np
import torch

from petastorm_amd.errors import NoDataAvailableError
from petastorm_amd.etl import dataset_metadata as dsm
from petastorm_amd.gpu.decoder import ByteArrayColumn, GpuRowGroupDecoder
from petastorm_amd.gpu.hbm_cache import HbmCache
from petastorm_amd.ngram import NGram
from petastorm_amd.parallel import epochs as epoch_sync
from petastorm_amd.transform import transform_schema
from petastorm_amd.unischema import match_unischema_fields
from petastorm_amd.codecs import (CompressedImageCodec, NdarrayCodec,
                                  CompressedNdarrayCodec)

logger = logging.getLogger(__name__)

_PREFETCH_DEPTH = 2


class _Ended(object):
    """Placeholder for an exhausted IO queue in the round-robin scan."""

    def get(self):
        raise RuntimeError('queue already ended')


_ENDED_QUEUE = _Ended()


def _next_live(qs, rr):
    n = len(qs)
    for step in range(1, n + 1):
        cand = (rr + step) % n
        if qs[cand] is not _ENDED_QUEUE:
            return cand
    return rr


class _TraceRange(object):
    """rocTX range (torch.cuda.nvtx maps to roctx on ROCm) for rocprofv3
    --marker-trace; enabled with PSA_TRACE=1 (replaces the
    reference's per-worker cProfile as the pipeline-stage tracer)."""

    enabled = os.environ.get('PSA_TRACE') == '1'

    def __init__(self, name):
        self._name = name

    def __enter__(self):
        if self.enabled:
            torch.cuda.nvtx.range_push(self._name)
        return self

    def __exit__(self, *exc):
        if self.enabled:
            torch.cuda.nvtx.range_pop()


class _PinnedPool(object):
    """Reusable pinned host buffers (rounded up to 1 MiB steps)."""

    def __init__(self):
        self._free = {}
        self._lock = threading.Lock()

    def get(self, nbytes):
        size = ((nbytes + (1 << 20) - 1) >> 20) << 20
        with self._lock:
            lst = self._free.get(size)
            if lst:
                return lst.pop()
        return torch.empty(size, dtype=torch.uint8,
                           pin_memory=torch.cuda.is_available())

    def put(self, buf):
        with self._lock:
            self._free.setdefault(buf.numel(), []).append(buf)


class GpuBatchReader(object):
    def __init__(self, fs, path_or_paths, schema_fields=None,
                 shuffle_row_groups=True, shuffle_rows=False, predicate=None,
                 rowgroup_selector=None,
                 num_epochs=1, cur_shard=None, shard_count=None, seed=None,
                 transform_spec=None, filters=None, device='cuda',
                 cache_type=None, cache_size_limit=None, pipeline_depth=3,
                 io_threads=2, decode_streams=3):
        requested_ngram = None
        if isinstance(schema_fields, NGram):
            requested_ngram = schema_fields
            schema_fields = None  # resolved against the schema below
        self._fs = fs
        self._paths = path_or_paths
        self.device = torch.device(device)
        if self.device.type == 'cuda' and not torch.cuda.is_available():
            raise RuntimeError(
                "make_batch_reader(device='cuda') requires a GPU: "
                'torch.cuda.is_available() is False on this machine')
        self._decoder = GpuRowGroupDecoder(self.device)
        self._pin_pool = _PinnedPool()

        storage_schema, _ = dsm.infer_or_load_unischema(fs, path_or_paths)
        if requested_ngram is not None:
            # batched windowing over the decoded HBM columns replaces the
            # reference's per-row form_ngram loop (gpu/ngram.py; reference
            # ngram.py:225-270): each emitted item is {timestep: namedtuple}
            # with [n_windows, ...] tensor fields
            requested_ngram.resolve_regex_field_names(storage_schema)
            self._view_schema = storage_schema.create_schema_view(
                requested_ngram.get_field_names_at_all_timesteps())
        elif schema_fields is not None:
            matched = match_unischema_fields(storage_schema, schema_fields)
            if not matched:
                raise ValueError('schema_fields matched nothing')
            self._view_schema = storage_schema.create_schema_view(matched)
        else:
            self._view_schema = storage_schema
        self._storage_schema = storage_schema
        self.transform_spec = transform_spec
        self.schema = transform_schema(self._view_schema, transform_spec) \
            if transform_spec else self._view_schema
        # FusedImageNormalize transforms fuse into the jpeg color kernel
        # epilogue (transform.py); the callable detects the fused output
        # at emit time and passes it through
        from petastorm_amd.transform import FusedImageNormalize
        f = getattr(transform_spec, 'func', None)
        self._fused_image_norm = f if isinstance(f, FusedImageNormalize) \
            else None
        self.batched_output = True
        self.ngram = requested_ngram
        self.last_row_consumed = False

        self._pieces = dsm.load_row_groups(fs, path_or_paths)
        if filters:
            self._pieces = dsm.select_pieces_by_filters(fs, self._pieces,
                                                        filters)
        if rowgroup_selector is not None:
            # rowgroup-index selectors (reference reader.py:599-618)
            from petastorm_amd.etl.rowgroup_indexing import \
                load_rowgroup_indexes
            index_dict = load_rowgroup_indexes(fs, path_or_paths)
            missing = [n for n in rowgroup_selector.select_index_names()
                       if n not in index_dict]
            if missing:
                raise ValueError(
                    'Indexes {} are not available in the dataset '
                    '(available: {})'.format(missing, sorted(index_dict)))
            chosen = rowgroup_selector.select_row_groups(index_dict)
            self._pieces = [p for p in self._pieces if p.index in chosen]
        if not self._pieces:
            raise NoDataAvailableError('Dataset has no row groups')
        cur_shard, shard_count = epoch_sync.shard_for_rank(cur_shard,
                                                           shard_count)
        self._cur_shard, self._shard_count = cur_shard, shard_count
        if shard_count is not None and shard_count > len(self._pieces):
            raise NoDataAvailableError(
                'Number of row-groups ({}) < shard_count ({})'
                .format(len(self._pieces), shard_count))
        # agree a base seed ONCE (rank-0 broadcast if distributed and the
        # user gave none): every epoch permutation is then derived locally,
        # keeping the data path collective-free (see parallel/epochs.py)
        self._seed = epoch_sync.agree_seed(seed) \
            if (shuffle_row_groups and
                (seed is not None or epoch_sync._dist() is not None)) \
            else seed
        self._shuffle_row_groups = shuffle_row_groups
        self._shuffle_rows = shuffle_rows
        self._predicate = predicate
        self._num_epochs = num_epochs

        self._cache = HbmCache(cache_size_limit) \
            if cache_type == 'hbm' and cache_size_limit else None
        self._inflight_hosts = []
        self._cnt_free = []  # pinned row-count scalars (async predicate)
        self._pipeline_depth = max(1, int(pipeline_depth))
        self._io_threads = max(1, int(io_threads))
        # row-groups are independent, so each one decodes on its own HIP
        # stream (round-robin): row-group N+1's Huffman/snappy waves overlap
        # N's IDCT/color/transform and fill the chip — a single in-order
        # stream serializes them (jpeg_huffman alone launches only ~390
        # waves on a 256-CU part).  Emit order is unchanged; the per-row-
        # group event (take_pending) is recorded on the owning stream.
        n_streams = max(1, int(decode_streams))
        if self.device.type == 'cuda' and torch.cuda.is_available():
            self._streams = [torch.cuda.Stream(device=self.device)
                             for _ in range(n_streams)]
        else:
            self._streams = None

        # per-file metadata handles (footer parse once per file)
        self._file_md = {}
        self._stopped = False
        self._rows_epoch = 0
        self._epoch = 0
        self._piece_pos = 0
        self._resume = None
        # stage timing (enabled with PSA_TIMING=1; reported in diagnostics)
        self._timing_enabled = os.environ.get('PSA_TIMING') == '1'
        self.stage_times = {'io_wait': 0.0, 'decode': 0.0, 'codec': 0.0,
                            'postprocess': 0.0, 'io_read': 0.0,
                            'io_parse': 0.0, 'flush': 0.0}
        self._gen = self._generate()

    # ------------------------------------------------------------------
    _FILE_MD_CACHE_CAP = 4096

    def _metadata(self, path):
        # Parsed footer metadata only — the file handle is closed here, and
        # the cache is bounded so huge multi-file datasets cannot grow it
        # without limit (the FileMetaData objects carry no fd).
        if path not in self._file_md:
            import pyarrow.parquet as pq
            with self._fs.open(path, 'rb') as f:
                pf = pq.ParquetFile(f)
                entry = (pf.metadata, pf.schema)
            if len(self._file_md) >= self._FILE_MD_CACHE_CAP:
                self._file_md.pop(next(iter(self._file_md)))
            self._file_md[path] = entry
        return self._file_md[path]

    def _epoch_pieces(self, epoch):
        n = len(self._pieces)
        perm = epoch_sync.epoch_permutation(
            n, epoch, self._seed, self._shuffle_row_groups)
        if self._shard_count is not None:
            if epoch_sync._dist() is not None:
                # Equal shards (DistributedSampler-style remainder drop)
                # keep per-rank epochs the same LENGTH, so explicit
                # epoch_stats() calls and checkpoint cursors line up across
                # ranks; which groups are dropped rotates with the epoch
                # permutation, so coverage evens out across epochs.
                n_even = n - (n % self._shard_count)
                perm = perm[:n_even]
            perm = [p for pos, p in enumerate(perm)
                    if pos % self._shard_count == self._cur_shard]
        return [self._pieces[i] for i in perm]

    # ------------------------------------------------------------------
    def _io_worker(self, pieces, out_q):
        """Read + host-parse the given pieces in order, into ``out_q``.

        Several IO workers run concurrently (round-robin piece assignment,
        round-robin consumption) — a single thread's pread + native parse
        caps the pipeline at a few hundred row-groups/s on a contended host.
        Ordering stays deterministic because the consumer reads the
        per-thread queues in the same round-robin order.
        """
        all_columns = list(self._view_schema.fields.keys())
        try:
            for piece in pieces:
                if self._stopped:
                    break
                if self._cache is not None and \
                        self._cache_key(piece) in self._cache._store:
                    self._q_put(out_q, ('cached', piece, None, None))
                    continue
                # hive-partition fields live in the PATH, not the file:
                # keep them out of the physical read and materialize them
                # in _decode_piece (reference partitions= in piece.read,
                # arrow_reader_worker.py:358)
                parts = piece.partitions or {}
                columns = [c for c in all_columns if c not in parts]
                md, pschema = self._metadata(piece.path)
                t0 = time.perf_counter()
                if columns:
                    host, meta = self._decoder.read_rowgroup_bytes(
                        piece.path, md, pschema, piece.row_group, columns,
                        self._pin_pool)
                else:  # every requested field is a partition key
                    host = torch.empty(0, dtype=torch.uint8)
                    meta = {'num_rows': piece.num_rows, 'chunks': []}
                t1 = time.perf_counter()
                # host-only parse work (page walk, offset scans, image
                # headers) runs HERE so it overlaps GPU decode of other
                # row-groups
                plan = self._decoder.prepare_host(host, meta,
                                                  self._storage_schema)
                self.stage_times['io_read'] += t1 - t0
                self.stage_times['io_parse'] += time.perf_counter() - t1
                if not self._q_put(out_q, ('data', piece, host,
                                           (meta, plan))):
                    return
            self._q_put(out_q, ('end', None, None, None))
        except Exception as e:  # noqa: BLE001 - forwarded to consumer
            self._q_put(out_q, ('error', e, None, None))

    def _q_put(self, q, item):
        """Stop-aware bounded put: a blocked IO thread must wake when the
        reader stops, or process exit can finalize the interpreter while
        the thread is inside GIL-released native code (observed as a
        SIGABRT at teardown with slow host decompression in flight)."""
        while not self._stopped:
            try:
                q.put(item, timeout=0.1)
                return True
            except queue.Full:
                continue
        return False

    @staticmethod
    def _cache_key(piece):
        return '{}:{}'.format(piece.path, piece.row_group)

    def _reclaim_hosts(self):
        """Return pinned buffers whose H2D copies have completed."""
        still = []
        for ev, buf in self._inflight_hosts:
            if ev.query():
                self._pin_pool.put(buf)
            else:
                still.append((ev, buf))
        self._inflight_hosts = still

    # ------------------------------------------------------------------
    def _generate(self):
        epoch = 0
        skip = 0
        if self._resume is not None:
            epoch = self._resume.get('epoch', 0)
            skip = self._resume.get('piece_pos', 0)
            self._resume = None
        while self._num_epochs is None or epoch < self._num_epochs:
            self._epoch = epoch
            pieces = self._epoch_pieces(epoch)
            if skip:
                pieces = pieces[skip:]
                self._piece_pos = skip
                skip = 0
            else:
                self._piece_pos = 0
            n_io = min(self._io_threads, max(1, len(pieces)))
            qs = [queue.Queue(maxsize=_PREFETCH_DEPTH) for _ in range(n_io)]
            threads = [
                threading.Thread(target=self._io_worker,
                                 args=(pieces[t::n_io], qs[t]), daemon=True)
                for t in range(n_io)]
            self._live_io_threads = threads
            for t in threads:
                t.start()
            self._rows_epoch = 0
            # software pipeline: decode of the next row-groups is LAUNCHED
            # before earlier batches are yielded, keeping `pipeline_depth`
            # row-groups of H2D + kernels in flight across the stream pool
            # (fills the chip when a single row-group's decode launches few
            # waves); each row-group's pinned status verdict was copied at
            # dispatch, so emit only waits that row-group's event
            from collections import deque
            pending = deque()
            dispatched = self._piece_pos  # pieces fully processed so far

            def emit(entry):
                piece, columns, pmeta, snap, pos = entry
                t3 = time.perf_counter()
                self._decoder.check_and_recycle(snap)
                self.stage_times['flush'] += time.perf_counter() - t3
                if columns is None:
                    return None
                if self._streams:
                    # tensors were produced on the row-group's own stream;
                    # the event wait above proves the DATA ready, but the
                    # caching allocator must also not hand their blocks to
                    # that stream before the consumer stream is done
                    cur = torch.cuda.current_stream(self.device)
                    for v in columns.values():
                        if isinstance(v, torch.Tensor) and v.is_cuda:
                            v.record_stream(cur)
                t3 = time.perf_counter()
                batch = self._postprocess_emit(piece, columns, pmeta)
                self.stage_times['postprocess'] += time.perf_counter() - t3
                if batch is None:
                    return None
                if self.ngram is not None:
                    from petastorm_amd.gpu.ngram import form_ngram_batched
                    windows = form_ngram_batched(batch, self.ngram)
                    first = windows[min(windows)]
                    nwin = len(next(iter(first.values()))) if first else 0
                    if nwin == 0:
                        return None
                    nt = self.ngram.make_namedtuple(self.schema, windows)
                    self._piece_pos = pos
                    self._rows_epoch += nwin
                    return nt
                nt = self.schema.make_namedtuple(**batch)
                self._piece_pos = pos
                self._rows_epoch += len(nt[0])
                return nt

            rr = 0
            ended = 0
            while ended < n_io:
                t0 = time.perf_counter()
                while True:  # stop-aware get (mirrors _q_put)
                    if self._stopped:
                        return
                    try:
                        kind, piece, host, meta = qs[rr].get(timeout=0.2)
                        break
                    except queue.Empty:
                        continue
                self.stage_times['io_wait'] += time.perf_counter() - t0
                if kind == 'end':
                    ended += 1
                    qs[rr] = _ENDED_QUEUE
                    rr = _next_live(qs, rr)
                    continue
                if kind == 'error':
                    raise piece
                rr = _next_live(qs, rr)
                stream_ctx = (
                    torch.cuda.stream(
                        self._streams[dispatched % len(self._streams)])
                    if self._streams else contextlib.nullcontext())
                with stream_ctx:
                    entry = self._dispatch_one(piece, kind, host, meta)
                dispatched += 1
                pending.append(entry + (dispatched,))
                while len(pending) >= self._pipeline_depth:
                    out = emit(pending.popleft())
                    if out is not None:
                        yield out
            while pending:
                out = emit(pending.popleft())
                if out is not None:
                    yield out
            for t in threads:
                t.join()
            epoch += 1

    def _dispatch_one(self, piece, kind, host, meta):
        """Decode + dispatch-phase postprocess of one row-group on the
        CURRENT stream; returns a pending-pipeline entry (sans position)."""
        if kind == 'cached':
            columns = self._cache._store[self._cache_key(piece)]
            self._cache.get(self._cache_key(piece), lambda: columns)
        else:
            meta, plan = meta
            if self._cache is not None:
                columns = self._cache.get(
                    self._cache_key(piece),
                    lambda: self._decode_piece(piece, host, meta, plan))
            else:
                columns = self._decode_piece(piece, host, meta, plan)
            if host is not None:
                # the async H2D of `host` may still be in flight; reclaim
                # the pinned buffer only after an event recorded behind it
                # (on this row-group's stream) completes
                ev = torch.cuda.Event()
                ev.record()
                self._inflight_hosts.append((ev, host))
        self._reclaim_hosts()
        t2 = time.perf_counter()
        with _TraceRange('psa.postprocess'):
            cols, pmeta = self._postprocess(piece, columns)
        self.stage_times['postprocess'] += time.perf_counter() - t2
        return (piece, cols, pmeta, self._decoder.take_pending())

    # ------------------------------------------------------------------
    def _datetime_unit(self, path, name):
        """numpy datetime64 unit of a DATE/TIMESTAMP column ('D'/'ms'/'us'/
        'ns'), from the Parquet logical type; None for non-datetime."""
        _, pschema = self._metadata(path)
        for i in range(len(pschema)):
            col = pschema.column(i)
            if col.path.split('.')[0] != name:
                continue
            if col.physical_type == 'INT96':
                return 'ns'  # legacy Spark timestamps decode to int64 ns
            lt = str(col.logical_type).lower()
            if lt.startswith('date'):
                return 'D'
            if 'timestamp' in lt:
                for unit in ('nanoseconds', 'microseconds', 'milliseconds'):
                    if unit in lt:
                        return {'nanoseconds': 'ns', 'microseconds': 'us',
                                'milliseconds': 'ms'}[unit]
                return 'us'
            return None
        return None

    @staticmethod
    def _is_string_field(field):
        """Scalar str/bytes field (incl. ScalarCodec'd and schema-inferred
        string columns): eligible for the device string path."""
        if field is None or field.shape not in ((), None):
            return False
        dt = field.numpy_dtype
        if dt in (np.str_, np.bytes_):
            return True
        from decimal import Decimal
        if dt is Decimal:
            # stored as a string column (codecs._arrow_scalar_type); the
            # batch route surfaces decimals as strings on CPU too, so the
            # device string path gives route parity
            return True
        try:
            return np.dtype(dt).kind in 'SU'
        except TypeError:
            return False

    def _decode_piece(self, piece, host, meta, plan=None):
        t0 = time.perf_counter()
        with _TraceRange('psa.decode_rowgroup'):
            raw, dbuf = self._decoder.decode(host, meta,
                                             self._storage_schema, plan)
        self.stage_times['decode'] += time.perf_counter() - t0
        t0 = time.perf_counter()
        columns = {}
        assist = []
        for name, col in raw.items():
            field = self._storage_schema.fields.get(name)
            if col is None:
                assist.append(name)
                continue
            if isinstance(col, ByteArrayColumn):
                decoded = None
                codec = field.codec if field is not None else None
                if isinstance(codec, CompressedImageCodec) and \
                        codec.image_codec == 'jpeg':
                    fused = (self._fused_image_norm
                             if self._fused_image_norm is not None and
                             self._fused_image_norm.field == name else None)
                    decoded = self._decoder.decode_jpeg_column(
                        col, field, fused_norm=fused)
                elif isinstance(codec, CompressedImageCodec):
                    decoded = self._decoder.decode_png_column(col, field)
                elif isinstance(codec, CompressedNdarrayCodec):
                    decoded = self._decoder.decode_compressed_ndarray_column(
                        col, field)
                elif isinstance(codec, NdarrayCodec) or (
                        codec is None and field is not None and
                        field.shape not in ((), None)):
                    decoded = self._decoder.decode_ndarray_column(col, field)
                elif self._is_string_field(field):
                    # strings/raw binary: device page decode + boundary
                    # materialization (no pyarrow re-read)
                    decoded = self._decoder.decode_string_column(col, field)
                if decoded is None:
                    assist.append(name)
                else:
                    columns[name] = decoded
            else:
                dt = field.numpy_dtype if field is not None else None
                if isinstance(col, torch.Tensor):
                    if dt is np.datetime64:
                        # DATE/TIMESTAMP: physical ints decoded on device;
                        # materialize numpy datetime64 at the boundary
                        # (torch has no datetime dtype — CPU-route type
                        # parity, reference unischema.py:467-502 mapping).
                        # Null slots carry the int-min sentinel -> NaT.
                        unit = self._datetime_unit(piece.path, name) or 'us'
                        arr = col.cpu().numpy()
                        nat = arr == np.iinfo(arr.dtype).min
                        col = arr.astype('datetime64[{}]'.format(unit))
                        if nat.any():
                            col[nat] = np.datetime64('NaT')
                    # unsigned logical types over signed physical storage:
                    # same widening convention as the ndarray codec path
                    # (reference pytorch.py:40-70 sanitization)
                    elif dt is np.uint8:
                        col = col.to(torch.uint8)
                    elif dt is np.int8:
                        col = col.to(torch.int8)
                    elif dt is np.int16:
                        col = col.to(torch.int16)
                    elif dt is np.uint16:
                        col = col.to(torch.int32) & 0xFFFF
                    elif dt is np.uint32:
                        col = col.to(torch.int64) & 0xFFFFFFFF
                    elif dt is np.uint64:
                        # no torch uint64: numpy at the boundary
                        col = col.cpu().numpy().view(np.uint64)
                columns[name] = col
        if assist:
            columns.update(self._cpu_assist(piece, assist))
        # materialize requested hive-partition fields as constant columns
        # (CPU path: workers/batch_worker.py partition handling)
        parts = piece.partitions or {}
        n_rows = meta['num_rows'] if isinstance(meta, dict) else None
        for name in self._view_schema.fields:
            if name in columns or name not in parts:
                continue
            field = self._storage_schema.fields.get(name)
            raw = parts[name]
            try:
                val = np.dtype(field.numpy_dtype).type(raw) \
                    if field is not None else raw
                columns[name] = self._decoder._up(
                    np.full(n_rows, val))
            except (TypeError, ValueError):
                columns[name] = np.full(n_rows, str(raw), dtype=object)
        # NB: flush_status is called by the pipeline loop just before this
        # row-group's batch is yielded, so the sync overlaps decode of the
        # next row-group
        self.stage_times['codec'] += time.perf_counter() - t0
        return columns

    def _cpu_assist(self, piece, names):
        """CPU decode + upload for columns outside the GPU fast path."""
        from petastorm_amd.workers.batch_worker import \
            arrow_table_to_numpy_dict
        import pyarrow.parquet as pq
        with self._fs.open(piece.path, 'rb') as f:
            table = pq.ParquetFile(f).read_row_group(
                piece.row_group, columns=names)
        np_dict = arrow_table_to_numpy_dict(table, self._storage_schema, True)
        out = {}
        for k, v in np_dict.items():
            if isinstance(v, np.ndarray) and v.dtype.kind in 'iufb':
                out[k] = self._decoder._up(np.ascontiguousarray(v))
            else:
                out[k] = v  # strings/objects stay host-side
        return out

    # ------------------------------------------------------------------
    def _postprocess(self, piece, columns):
        """Dispatch-time half of postprocessing.  The predicate compaction
        is enqueued fully ASYNC: a stable argsort of the (negated) mask
        permutes kept rows to the front while the row count lands in pinned
        memory behind this row-group's kernels; ``_postprocess_emit`` does
        the final slice after the pipeline's event wait.  ``torch.nonzero``
        here would sync the stream and serialize the whole pipeline
        (measured 21 ms/row-group on the ngram config).

        Returns ``(columns, pinned_count_or_None)``; ``(None, None)`` for
        empty row-groups."""
        columns = dict(columns)
        n = 0
        for v in columns.values():
            n = len(v)
            break
        if not n:
            return None, None
        host_cnt = None
        if self._predicate is not None:
            mask = self._predicate_mask(columns)
            if mask is not None:
                if isinstance(mask, torch.Tensor) and mask.is_cuda and \
                        all(isinstance(v, torch.Tensor)
                            for v in columns.values()):
                    host_cnt = (self._cnt_free.pop() if self._cnt_free
                                else torch.empty((), dtype=torch.int64,
                                                 pin_memory=True))
                    host_cnt.copy_(mask.sum(), non_blocking=True)
                    order = torch.argsort(
                        mask.logical_not().to(torch.uint8), stable=True)
                    columns = {k: v.index_select(0, order)
                               for k, v in columns.items()}
                else:  # host mask / non-tensor columns: sync path
                    idx = torch.nonzero(mask, as_tuple=False).squeeze(1)
                    if idx.numel() == 0:
                        return None, None
                    columns = {k: (v.index_select(0, idx)
                                   if isinstance(v, torch.Tensor)
                                   else v[idx.cpu().numpy()])
                               for k, v in columns.items()}
        return columns, host_cnt

    def _postprocess_emit(self, piece, columns, host_cnt):
        """Emit-time half: slice the predicate-kept rows (count now valid
        — check_and_recycle waited the event), then shuffle/transform."""
        if host_cnt is not None:
            k = int(host_cnt.item())
            self._cnt_free.append(host_cnt)
            if k == 0:
                return None
            columns = {name: v[:k] for name, v in columns.items()}
        if self._shuffle_rows:
            n2 = len(next(iter(columns.values())))
            g = None
            if self._seed is not None:
                g = torch.Generator(device='cpu')
                g.manual_seed((self._seed + piece.index) % (2 ** 31))
            perm = self._decoder._up(torch.randperm(n2, generator=g))
            columns = {k: (v.index_select(0, perm)
                           if isinstance(v, torch.Tensor)
                           else v[perm.cpu().numpy()])
                       for k, v in columns.items()}
        if self.transform_spec is not None:
            if self.transform_spec.func:
                columns = self.transform_spec.func(columns)
            keep = set(self.schema.fields.keys())
            columns = {k: v for k, v in columns.items() if k in keep}
        return columns

    def _predicate_mask(self, columns):
        names = list(self._predicate.get_fields())
        n = len(next(iter(columns.values())))
        # 1) fully on-device: most predicates (in_lambda with tensor-
        #    compatible expressions, in_set via torch.isin) evaluate
        #    directly on the CUDA columns
        try:
            dev_cols = {f: columns[f] for f in names}
            if all(isinstance(v, torch.Tensor) for v in dev_cols.values()):
                res = self._predicate.do_include(dev_cols)
                if isinstance(res, torch.Tensor) and \
                        res.dtype == torch.bool and res.numel() == n:
                    return res
        except Exception:  # noqa: BLE001 - fall back to host evaluation
            pass
        # 2) host fallback: numpy vectorized path, mask uploaded
        host_cols = {}
        for f in names:
            v = columns[f]
            host_cols[f] = v.cpu().numpy() if isinstance(v, torch.Tensor) \
                else np.asarray(v)
        mask = self._predicate.do_include_vectorized(host_cols)
        return self._decoder._up(np.asarray(mask, dtype=bool))

    # ------------------------------------------------------------------
    def __iter__(self):
        return self

    def __next__(self):
        if self._stopped:
            raise StopIteration
        try:
            return next(self._gen)
        except StopIteration:
            self.last_row_consumed = True
            raise

    next = __next__

    def reset(self):
        self._gen = self._generate()
        self.last_row_consumed = False

    # ------------------------------------------------------------------
    # iterator-state checkpointing (the reference offers
    # only determinism-by-seed as a resume substitute; this framework saves
    # the cursor explicitly).  Requires a seeded reader (or an initialized
    # process group, where rank 0's broadcast makes epochs reproducible
    # within one job but NOT across restarts — so persist with seed set).
    # ------------------------------------------------------------------
    def epoch_stats(self):
        """All-gather per-rank rows consumed in the current epoch.

        COLLECTIVE: call from the application at a point where every rank
        calls it together (e.g. the training loop's epoch boundary) — it is
        deliberately not part of the reader's generator, whose progress is
        consumption-driven and not synchronized across ranks
        (parallel/epochs.py module docstring)."""
        return epoch_sync.epoch_end_sync(self._rows_epoch)

    def state_dict(self):
        return {'epoch': self._epoch, 'piece_pos': self._piece_pos,
                'seed': self._seed}

    def load_state_dict(self, state):
        if state.get('seed') != self._seed:
            raise ValueError('state was saved with seed={!r}; reader has '
                             'seed={!r} — epoch permutations would diverge'
                             .format(state.get('seed'), self._seed))
        self._resume = {'epoch': state['epoch'],
                        'piece_pos': state['piece_pos']}
        self._gen = self._generate()
        self.last_row_consumed = False

    def stop(self):
        self._stopped = True

    def join(self):
        # IO threads must be OUT of native code before interpreter exit
        for t in getattr(self, '_live_io_threads', []):
            t.join(timeout=10.0)
        if self._cache is not None:
            self._cache.cleanup()

    @property
    def diagnostics(self):
        d = {'cpu_assist_columns': sorted(self._decoder.cpu_assist_columns),
             'stage_times': dict(self.stage_times),
             'staging_allocs': self._decoder.staging_allocs,
             'staging_copy_s': round(self._decoder.staging_copy_s, 4),
             'staging_copy_by_key': sorted(
                 self._decoder.staging_copy_by_key.items(),
                 key=lambda kv: -kv[1][1])[:5]}
        if self._cache is not None:
            d.update(hbm_cache_hits=self._cache.hits,
                     hbm_cache_misses=self._cache.misses,
                     hbm_cache_bytes=self._cache.size_bytes)
        return d

    def __enter__(self):
        return self

    def __exit__(self, exc_type, exc_val, exc_tb):
        self.stop()
        self.join()
