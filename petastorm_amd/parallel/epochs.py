"""Multi-GPU epoch coordination over torch.distributed (RCCL on ROCm).

Replaces the reference's "same seed everywhere" convention for sharded
readers (reference petastorm/reader.py:573-597 — each rank permutes row
groups with an identical seed and takes ``index % shard_count == cur_shard``)
with a design that is collective-free on the data path:

* reader construction: when no user seed is given, rank 0 samples a base
  seed and broadcasts it ONCE (``agree_seed`` — all ranks construct the
  reader together, so this is the only point where lock-step is guaranteed).
* every epoch's row-group permutation is then derived LOCALLY from
  (seed, epoch) — identical on every rank with zero messages.
* per-rank epoch statistics (``epoch_end_sync``) are an EXPLICIT
  application-called collective, not part of the reader's generator.

Why not a broadcast/all-gather at every epoch boundary (the first design)?
Ranks do not cross epoch boundaries after the same number of consumer
steps once anything makes per-row-group yields uneven — a predicate that
filters different row counts per shard, shuffle-buffer thresholds, or
uneven storage.  A rank blocking in a boundary collective that its peers
will only reach k steps later (or never, if their step budget runs out
first) deadlocks the job; collectives inside a demand-driven generator are
ordered by CONSUMPTION, which nothing synchronizes.  The data path is
therefore collective-free by construction, matching the reference's
convention while fixing its "hope the seeds match" weakness via the
init-time agreement.

Every function degrades gracefully to single-process semantics when
torch.distributed is not initialized, and runs on the gloo backend for
CPU-only tests (world_size > 1 multi-process tests run here without a GPU).
"""

import numpy as np


def _dist():
    import torch.distributed as dist
    return dist if dist.is_available() and dist.is_initialized() else None


def agree_seed(seed=None):
    """A base seed identical on every rank.

    A user-provided seed is already rank-consistent and returned as-is.
    Otherwise rank 0 samples one and broadcasts it — called ONCE from
    reader construction, the only naturally lock-step point.  Without a
    process group, a locally sampled seed is returned.
    """
    import torch
    if seed is not None:
        return int(seed)
    dist = _dist()
    local = int(np.random.randint(0, 2 ** 31 - 1))
    if dist is None:
        return local
    device = _collective_device(dist)
    t = torch.tensor([local if dist.get_rank() == 0 else 0],
                     dtype=torch.int64, device=device)
    dist.broadcast(t, src=0)
    return int(t.cpu().item())


def epoch_permutation(n_items, epoch, seed=None, shuffle=True):
    """The epoch's row-group order — derived locally from (seed, epoch) so
    every rank computes the same order with no communication.

    ``seed=None`` gives an unseeded local permutation and is only meaningful
    single-process; distributed readers pass a seed from :func:`agree_seed`.
    """
    if not shuffle:
        return np.arange(n_items, dtype=np.int64)
    rng = np.random.RandomState(
        None if seed is None else (seed + epoch) % (2 ** 31))
    return rng.permutation(n_items).astype(np.int64)


def epoch_end_sync(rows_consumed):
    """All-gather per-rank consumed-row counts + barrier.

    COLLECTIVE — every rank must call it together.  It is deliberately NOT
    called from the reader's generator (see module docstring); applications
    call it (or ``GpuBatchReader.epoch_stats()``) at points where all ranks
    are known to be synchronized, e.g. a training-loop epoch boundary.

    Returns the list of per-rank counts (len == world_size), or
    ``[rows_consumed]`` when not distributed.
    """
    import torch
    dist = _dist()
    if dist is None:
        return [int(rows_consumed)]
    device = _collective_device(dist)
    t = torch.tensor([int(rows_consumed)], dtype=torch.int64, device=device)
    out = [torch.zeros_like(t) for _ in range(dist.get_world_size())]
    dist.all_gather(out, t)
    dist.barrier()
    return [int(x.item()) for x in out]


def _collective_device(dist):
    import torch
    backend = dist.get_backend()
    if 'nccl' in str(backend) and torch.cuda.is_available():
        return torch.device('cuda', torch.cuda.current_device())
    return torch.device('cpu')


def shard_for_rank(cur_shard=None, shard_count=None):
    """Default the shard spec to the process-group rank/size when present."""
    dist = _dist()
    if cur_shard is not None or dist is None:
        return cur_shard, shard_count
    return dist.get_rank(), dist.get_world_size()
