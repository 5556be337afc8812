"""Multi-GPU sharding of the label index space.

The reference already partitions the label space by file
(InitOpts.MaxFileSize, activation/post.go:56-60,166-182); across the 8 GPUs
of one MI355X node the same axis shards embarrassingly: contiguous index
ranges, one engine session (and postdata file set) per GPU, no data-path
collective.  The ONE exchange is the VRF nonce: each shard tracks its local
minimum full label (common/types/activation.go:311-313) and the global nonce
is an 8-way min-reduce of (label[32], index) pairs — 40 bytes per rank,
latency-only.  Implemented as all_gather over
torch.distributed (RCCL on GPUs, gloo in CPU tests) + a deterministic local
reduction, so every rank ends with the identical winner.
"""
from __future__ import annotations

from typing import List, Optional, Tuple


def shard_range(total_labels: int, world_size: int,
                rank: int) -> Tuple[int, int]:
    """Contiguous near-equal [start, end) shard for `rank`.  The union over
    ranks is exactly [0, total_labels) with no overlap."""
    base = total_labels // world_size
    rem = total_labels % world_size
    start = rank * base + min(rank, rem)
    end = start + base + (1 if rank < rem else 0)
    return start, end


def merge_nonces(cands: List[Optional[Tuple[int, bytes]]]
                 ) -> Optional[Tuple[int, bytes]]:
    """Deterministic min-reduce of per-shard VRF candidates.
    Each candidate is (index, full_label[32]); the winner has the smallest
    label (big-endian lexicographic), ties broken by smallest index — the
    same ordering the initializer itself uses."""
    best = None
    for c in cands:
        if c is None:
            continue
        if best is None or (c[1], c[0]) < (best[1], best[0]):
            best = c
    return best


def allreduce_nonce(local: Optional[Tuple[int, bytes]],
                    group=None) -> Optional[Tuple[int, bytes]]:
    """Global VRF nonce across ranks via one all_gather_object (the payload
    is 40 bytes per rank; bandwidth is irrelevant — SURVEY.md §5)."""
    import torch.distributed as dist
    if not dist.is_initialized() or dist.get_world_size(group) == 1:
        return local
    gathered: List[Optional[Tuple[int, bytes]]] = \
        [None] * dist.get_world_size(group)
    dist.all_gather_object(gathered, local, group=group)
    return merge_nonces(gathered)
