"""Sharded multi-GPU index operations.

SURVEY.md §2.2 P4/P7: each rank holds a shard of the library; fan-out
queries scan every shard and merge per-query top-k via all-gather; index
builds train the coarse quantizer cooperatively (RCCL all-reduce on
centroid partials, ops/kmeans.py group arg) and all-gather the
embedding shards. xGMI note (§5.8): these payloads are small (top-k
rows, centroid sums), so latency dominates — single collectives, no
bucketing games.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.distributed as dist

from audiomuse_amd.index.ivf import IVFIndex


def shard_bounds(n: int, world: int, r: int) -> Tuple[int, int]:
    base = n // world
    extra = n % world
    start = r * base + min(r, extra)
    return start, start + base + (1 if r < extra else 0)


def sharded_topk_query(index: IVFIndex, q: torch.Tensor, k: int,
                       nprobe: Optional[int] = None,
                       group=None) -> Tuple[torch.Tensor, torch.Tensor]:
    """Every rank queries ITS shard index; results merge to identical
    (dist, global_id) top-k on all ranks via all_gather.

    index.ids must already hold GLOBAL ids for this rank's shard.
    Returns (Q, k) dists + ids, same on every rank.
    """
    local_d, local_i = index.query(q, k, nprobe=nprobe)
    if local_d.dim() == 1:
        local_d = local_d.unsqueeze(0)
        local_i = local_i.unsqueeze(0)
    if group is None and not dist.is_initialized():
        return local_d, local_i
    world = dist.get_world_size(group)
    gather_d = [torch.empty_like(local_d) for _ in range(world)]
    gather_i = [torch.empty_like(local_i) for _ in range(world)]
    dist.all_gather(gather_d, local_d.contiguous(), group=group)
    dist.all_gather(gather_i, local_i.contiguous(), group=group)
    all_d = torch.cat(gather_d, dim=1)
    all_i = torch.cat(gather_i, dim=1)
    top = torch.topk(all_d, min(k, all_d.shape[1]), dim=1, largest=False)
    return top.values, all_i.gather(1, top.indices)


def build_sharded_index(local_vectors: torch.Tensor,
                        global_ids: torch.Tensor, *,
                        metric: str = "angular",
                        storage: Optional[str] = None,
                        nlist: Optional[int] = None,
                        device: str = "cpu", seed: int = 0,
                        group=None) -> IVFIndex:
    """Cooperative build: the k-means quantizer trains over ALL shards
    (gradient of SURVEY P7 — all-reduced centroid partials); each rank
    packs only its own shard's cells. Queries then use
    sharded_topk_query for the global answer."""
    g = group if (group is not None or dist.is_initialized()) else None
    return IVFIndex.build(local_vectors, ids=global_ids, metric=metric,
                          storage=storage, nlist=nlist, device=device,
                          seed=seed, group=g)


def allgather_embeddings(local: torch.Tensor, group=None) -> torch.Tensor:
    """Variable-length all-gather of embedding shards (index build
    replication path)."""
    if not dist.is_initialized():
        return local
    world = dist.get_world_size(group)
    # Count exchange must ride the same device as the payload: with the
    # nccl/RCCL backend (the GPU deployment shape) collectives on CPU
    # tensors fail outright.
    n_local = torch.tensor([local.shape[0]], dtype=torch.long,
                           device=local.device)
    counts = [torch.zeros(1, dtype=torch.long, device=local.device)
              for _ in range(world)]
    dist.all_gather(counts, n_local, group=group)
    maxn = int(max(c.item() for c in counts))
    padded = torch.zeros(maxn, local.shape[1], dtype=local.dtype,
                         device=local.device)
    padded[: local.shape[0]] = local
    gathered = [torch.empty_like(padded) for _ in range(world)]
    dist.all_gather(gathered, padded, group=group)
    return torch.cat([gathered[r][: int(counts[r].item())]
                      for r in range(world)], dim=0)
