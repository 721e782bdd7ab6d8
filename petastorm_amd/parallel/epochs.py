"""Multi-GPU epoch coordination over torch.distributed (RCCL on ROCm).

Replaces the reference's "same seed everywhere" convention for sharded
readers (reference petastorm/reader.py:573-597 — each rank permutes row
groups with an identical seed and takes ``index % shard_count == cur_shard``)
with an explicit collective when a process group exists:

* epoch start: rank 0 samples the epoch's row-group permutation and
  broadcasts it (1 small message over xGMI), so sharding is consistent even
  without a user-provided seed.
* epoch end: all-gather of per-rank consumed row counts + barrier, replacing
  the reference's implicit per-rank ventilator completion
  (reference ventilator.py:124-126).

Every function degrades gracefully to single-process semantics when
torch.distributed is not initialized, and runs on the gloo backend for
CPU-only tests (world_size > 1 multi-process tests run here without a GPU).

Collective-ordering contract: every rank must execute the same sequence of
collectives.  GpuBatchReader guarantees it by dropping the per-epoch
row-group remainder so all shards are equal-sized (readers on different
ranks then hit epoch_permutation / epoch_end_sync in lock-step regardless of
relative progress within an epoch).
"""

import numpy as np


def _dist():
    import torch.distributed as dist
    return dist if dist.is_available() and dist.is_initialized() else None


def epoch_permutation(n_items, epoch, seed=None, shuffle=True):
    """The epoch's row-group order, identical on every rank.

    With a process group: rank 0 samples (seeded or not) and broadcasts.
    Without: seeded local RNG (reference behavior).
    """
    import torch
    if not shuffle:
        return np.arange(n_items, dtype=np.int64)
    dist = _dist()
    if dist is None:
        rng = np.random.RandomState(
            None if seed is None else (seed + epoch) % (2 ** 31))
        return rng.permutation(n_items).astype(np.int64)
    if dist.get_rank() == 0:
        rng = np.random.RandomState(
            None if seed is None else (seed + epoch) % (2 ** 31))
        perm = torch.from_numpy(rng.permutation(n_items).astype(np.int64))
    else:
        perm = torch.empty(n_items, dtype=torch.int64)
    device = _collective_device(dist)
    perm = perm.to(device)
    dist.broadcast(perm, src=0)
    return perm.cpu().numpy()


def epoch_end_sync(rows_consumed):
    """All-gather per-rank consumed-row counts + barrier at an epoch boundary.

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
