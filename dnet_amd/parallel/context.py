"""Context (sequence) parallelism for decode attention.

For contexts too long for one GPU's HBM (or to spread KV-read bandwidth),
the KV cache shards along the SEQUENCE axis across the ranks of a group:
rank r holds positions [r*shard, (r+1)*shard). Each rank computes the
unnormalized flash-decode partials (acc, m, l) over its local positions —
the same combinable form the split-S kernel uses within one GPU — then the
partials are all-gathered (tiny: Hq*(Dv+2) floats per sequence) and merged
locally. Numerically identical to full attention: max/sum-exp merging is
associative across shards.

This is the designed-in long-context substrate SURVEY.md §5 calls for
(the reference has none — "Long context 🚧" README.md:51); the ring
executor wires it up when a stage's KV exceeds one GPU (round-2
integration; the building block is tested CPU (gloo) + GPU here).
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist

from .. import ops


def shard_bounds(total_len: int, world: int, rank: int) -> tuple[int, int]:
    """Contiguous S-axis shard [s0, s1) for this rank (remainder to the
    last rank)."""
    shard = total_len // world
    s0 = rank * shard
    s1 = total_len if rank == world - 1 else s0 + shard
    return s0, s1


def local_lengths(pos: torch.Tensor, cap: int, rank: int) -> torch.Tensor:
    """Per-sequence valid length on this rank for capacity-block sharding
    (rank r owns global positions [r*cap, (r+1)*cap))."""
    return (pos - rank * cap).clamp(0, cap).to(torch.int32)


def cp_attn_decode(q: torch.Tensor, kcache: torch.Tensor,
                   vcache: torch.Tensor, pos_local: torch.Tensor,
                   scale: float, group=None,
                   sinks: Optional[torch.Tensor] = None,
                   kscale: Optional[torch.Tensor] = None,
                   vscale: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Decode attention over a sequence-sharded KV cache.

    q [B, Hq, D] (replicated across the group), kcache/vcache this rank's
    shard, ``pos_local`` [B] int32 = number of valid positions in the local
    shard. Returns out [B, Hq, Dv], identical on every rank.
    """
    parts = ops.attn_decode_partials(q, kcache, vcache, pos_local, scale,
                                     kscale, vscale)      # [B,Hq,s,Dv+2]
    world = dist.get_world_size(group) if dist.is_initialized() else 1
    if world > 1:
        gathered = [torch.empty_like(parts) for _ in range(world)]
        dist.all_gather(gathered, parts.contiguous(), group=group)
        parts = torch.cat(gathered, dim=2)
    return ops.attn_combine(parts, sinks)
