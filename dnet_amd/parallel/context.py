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


def _partials_windowed(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                       ln: torch.Tensor, scale: float, base: int,
                       pos: torch.Tensor, window: int) -> torch.Tensor:
    """Decode partials over a sequence shard with a sliding window.

    q [B, Hq, D]; k/v this rank's shard [B, Hkv, cap, D(v)]; ln [B] valid
    local length; base = global position of local slot 0; pos [B] global
    query positions. Returns [B, Hq, 1, Dv+2] f32 (acc, m, l) — same
    combinable form as the split-S kernel. Torch ops (einsum + fp32
    reductions); the windowless fast path stays on the hand kernel.
    """
    B, Hq, D = q.shape
    Hkv = k.shape[1]
    cap = k.shape[2]
    Dv = v.shape[-1]
    G = Hq // Hkv
    gpos = base + torch.arange(cap, device=q.device).view(1, -1)   # [1,cap]
    live = gpos < (base + ln.view(B, 1))                           # causal
    if window and window > 0:
        live &= gpos > (pos.view(B, 1) - window)
    qg = q.view(B, Hkv, G, D).float()
    s = torch.einsum("bhgd,bhsd->bhgs", qg, k.float()) * scale
    s = s.masked_fill(~live.view(B, 1, 1, cap), float("-inf"))
    m = s.amax(dim=-1)                                             # [B,Hkv,G]
    p = torch.exp(s - m.unsqueeze(-1))
    p = torch.nan_to_num(p, nan=0.0)        # fully-dead rows -> zeros
    l = p.sum(-1)
    acc = torch.einsum("bhgs,bhsd->bhgd", p, v.float())
    out = torch.empty(B, Hq, 1, Dv + 2, dtype=torch.float32,
                      device=q.device)
    out[..., 0, :Dv] = acc.reshape(B, Hq, Dv)
    out[..., 0, Dv] = m.reshape(B, Hq)
    out[..., 0, Dv + 1] = l.reshape(B, Hq)
    return out


def cp_attn_decode_windowed(q, kcache, vcache, pos, scale, cap, rank,
                            window, group=None, sinks=None, kscale=None,
                            vscale=None):
    """Sliding-window decode attention over a sequence-sharded cache
    (VERDICT r1 item 9 — window + CP). Ranks whose shard is entirely
    below the window contribute empty partials (m=-inf, l=0) that the
    combine ignores."""
    from ..ops import reference as _r
    k = kcache if kscale is None else _r.dequant_kv(kcache, kscale)
    v = vcache if vscale is None else _r.dequant_kv(vcache, vscale)
    ln = local_lengths(pos + 1, cap, rank)
    parts = _partials_windowed(q, k, v, ln, scale, rank * cap, pos.long(),
                               window)
    world = dist.get_world_size(group) if dist.is_initialized() else 1
    if world > 1:
        gathered = [torch.empty_like(parts) for _ in range(world)]
        dist.all_gather(gathered, parts.contiguous(), group=group)
        parts = torch.cat(gathered, dim=2)
    return ops.attn_combine(parts, sinks)


def cp_prefill_attention(q: torch.Tensor, kcache, vcache, pos_end,
                         q_pos0: int, scale: float, cap: int, rank: int,
                         window: int = 0, group=None, sinks=None,
                         kscale=None, vscale=None) -> torch.Tensor:
    """Prefill attention over a sequence-sharded KV cache WITHOUT
    gathering the full cache (the round-1 CP prefill
    transiently all-gathered the whole KV per layer, defeating CP's
    memory purpose). Each rank computes causal/windowed partials of the
    query chunk vs its LOCAL shard (transient memory: T x cap scores);
    the tiny (m, l, acc) partials are all-gathered and merged.

    q [B, Hq, T, D] (replicated); kcache/vcache local shards
    [B, Hkv, cap, D(v)]; pos_end = global sequence length including this
    chunk; q_pos0 = global position of q[:, :, 0]. Returns [B, Hq, T, Dv].
    """
    from ..ops import reference as _r
    k = kcache if kscale is None else _r.dequant_kv(kcache, kscale)
    v = vcache if vscale is None else _r.dequant_kv(vcache, vscale)
    B, Hq, T, D = q.shape
    Hkv = k.shape[1]
    G = Hq // Hkv
    Dv = v.shape[-1]
    ln = max(0, min(pos_end - rank * cap, cap))
    dev = q.device
    if ln == 0:
        parts = torch.full((B, Hq * T, 1, Dv + 2), 0.0, device=dev)
        parts[..., Dv] = float("-inf")
    else:
        gpos = rank * cap + torch.arange(ln, device=dev).view(1, -1)
        qpos = q_pos0 + torch.arange(T, device=dev).view(-1, 1)
        live = gpos <= qpos                                    # [T, ln]
        if window and window > 0:
            live &= gpos > qpos - window
        qg = q.view(B, Hkv, G, T, D).float()
        s = torch.einsum("bhgtd,bhsd->bhgts", qg, k[:, :, :ln].float())
        s = s * scale
        s = s.masked_fill(~live.view(1, 1, 1, T, ln), float("-inf"))
        m = s.amax(dim=-1)
        p = torch.nan_to_num(torch.exp(s - m.unsqueeze(-1)), nan=0.0)
        l = p.sum(-1)
        acc = torch.einsum("bhgts,bhsd->bhgtd", p, v[:, :, :ln].float())
        parts = torch.empty(B, Hq * T, 1, Dv + 2, dtype=torch.float32,
                            device=dev)
        parts[..., 0, :Dv] = acc.reshape(B, Hq * T, Dv)
        parts[..., 0, Dv] = m.reshape(B, Hq * T)
        parts[..., 0, Dv + 1] = l.reshape(B, Hq * T)
    world = dist.get_world_size(group) if dist.is_initialized() else 1
    if world > 1:
        gathered = [torch.empty_like(parts) for _ in range(world)]
        dist.all_gather(gathered, parts.contiguous(), group=group)
        parts = torch.cat(gathered, dim=2)
    sk = None
    if sinks is not None:
        sk = sinks.view(Hq, 1).expand(Hq, T).reshape(Hq * T)
    out = ops.attn_combine(parts, sk)                 # [B, Hq*T, Dv]
    return out.view(B, Hq, T, Dv)
