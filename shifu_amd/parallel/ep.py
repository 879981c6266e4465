"""Sharded (model-parallel) embeddings with all-to-all routing over xGMI.

The DP path replicates embedding arenas and allgathers sparse grads — at 8
ranks that wire volume is ~7x the lookup traffic and dominates the step.
This module shards the arena across ranks instead (row r lives on rank
r % world) and routes LOOKUPS, not parameters:

  forward : ids -> bucket by owner -> all_to_all(ids) -> owner gathers from
            its local shard (HIP gather kernel) -> all_to_all(values, bf16)
            -> unpermute -> [B, F*D]
  backward: reverse all_to_all of the output-grad rows; each owner receives
            the COMPLETE gradient for its rows -> sparse grad on the local
            shard (duplicate-tolerant atomic update kernels apply it) -> NO
            parameter synchronization at all.

Per rank per step this moves ~2 x n x D x 2 bytes point-to-point (n = B*F)
instead of (world-1) x n x D x 4 broadcast-style — the xGMI-native design
(7 p2p links per GPU).  This is the standard DLRM-style hybrid:
dense params stay data-parallel, embeddings are expert-parallel.

The wire dtype is bf16; owners accumulate in fp32.
gloo (CPU tests) lacks all_to_all, so a gather-based fallback emulates it —
the NCCL/RCCL path uses dist.all_to_all_single.
"""
from __future__ import annotations

import math
from typing import List, Sequence

import torch
import torch.distributed as dist

from shifu_amd.ops.dispatch import use_hip, hip_ops


def _all_to_all_single(out: torch.Tensor, inp: torch.Tensor,
                       out_splits: List[int], in_splits: List[int]) -> None:
    """dist.all_to_all_single with a gloo fallback (gloo has no all-to-all)."""
    backend = dist.get_backend()
    if backend != "gloo":
        dist.all_to_all_single(out, inp, out_splits, in_splits)
        return
    world = dist.get_world_size()
    rank = dist.get_rank()
    # emulate: every rank gathers every rank's input + split table, then
    # slices out its own chunks.  O(world * n) wire — test-only path.
    splits_t = torch.tensor(in_splits, dtype=torch.int64)
    all_splits = [torch.zeros_like(splits_t) for _ in range(world)]
    dist.all_gather(all_splits, splits_t)
    nmax = max(int(t.sum()) for t in all_splits)
    pad_shape = (nmax,) + tuple(inp.shape[1:])
    padded = torch.zeros(pad_shape, dtype=inp.dtype)
    padded[:inp.shape[0]] = inp
    gathered = [torch.zeros_like(padded) for _ in range(world)]
    dist.all_gather(gathered, padded)
    chunks = []
    for src in range(world):
        off = int(all_splits[src][:rank].sum())
        cnt = int(all_splits[src][rank])
        chunks.append(gathered[src][off:off + cnt])
    res = torch.cat(chunks) if chunks else inp.new_zeros((0,) + tuple(inp.shape[1:]))
    out.copy_(res)


class _EPGatherFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, arena_local: torch.Tensor, flat_ids: torch.Tensor,
                F: int, D: int, world: int, rank: int):
        n = flat_ids.numel()
        g = flat_ids.reshape(-1)
        owner = (g % world).to(torch.int64)
        # launch the split-size exchange FIRST so it overlaps the sort; read
        # the result back with ONE host transfer (per-element int() on device
        # tensors would be one sync each — 2*world syncs per lookup)
        send_counts = torch.bincount(owner, minlength=world)
        all_counts = [torch.zeros_like(send_counts) for _ in range(world)]
        dist.all_gather(all_counts, send_counts)
        perm = torch.argsort(owner, stable=True)
        ids_sorted = g[perm]
        cnt = torch.stack(all_counts).cpu()
        in_splits = [int(c) for c in cnt[rank]]                   # what I send
        out_splits = [int(cnt[src][rank]) for src in range(world)]
        m = sum(out_splits)

        recv_ids = torch.empty(m, dtype=g.dtype, device=g.device)
        _all_to_all_single(recv_ids, ids_sorted.contiguous(), out_splits, in_splits)

        local_rows = torch.div(recv_ids, world, rounding_mode="floor")
        if arena_local.dtype == torch.bfloat16 and use_hip(arena_local):
            vals = hip_ops().embedding_gather(arena_local, local_rows.reshape(-1, 1))
        else:
            vals = arena_local.index_select(0, local_rows)
        vals = vals.reshape(m, D)

        back = torch.empty(n, D, dtype=vals.dtype, device=vals.device)
        _all_to_all_single(back, vals.contiguous(), in_splits, out_splits)

        out = torch.empty_like(back)
        out[perm] = back
        ctx.save_for_backward(perm, local_rows)
        ctx.splits = (in_splits, out_splits)
        ctx.world, ctx.D = world, D
        ctx.arena_shape = arena_local.shape
        ctx.arena_dtype = arena_local.dtype
        B = n // F
        return out.reshape(B, F * D)

    @staticmethod
    def backward(ctx, dout: torch.Tensor):
        perm, local_rows = ctx.saved_tensors
        in_splits, out_splits = ctx.splits
        D, world = ctx.D, ctx.world
        n = perm.numel()
        dvals = dout.reshape(n, D)
        dsorted = dvals[perm].contiguous()
        grad_rows = torch.empty(local_rows.numel(), D, dtype=dsorted.dtype,
                                device=dsorted.device)
        _all_to_all_single(grad_rows, dsorted, out_splits, in_splits)
        # data-parallel loss averaging: global grad = mean over ranks
        grad_rows = grad_rows / world
        grad = torch.sparse_coo_tensor(local_rows.reshape(1, -1),
                                       grad_rows.to(ctx.arena_dtype),
                                       ctx.arena_shape)
        return grad, None, None, None, None, None


class _EPGatherPairFn(torch.autograd.Function):
    """Two arenas (same vocab layout, e.g. Wide&Deep's D=1 wide weights and
    D=64 deep vectors) share ONE routing: one sort, one split-size exchange,
    one id all-to-all, and a single combined value all-to-all of [m, D1+D2]."""

    @staticmethod
    def forward(ctx, arena1, arena2, flat_ids, F, D1, D2, world, rank):
        n = flat_ids.numel()
        g = flat_ids.reshape(-1)
        owner = (g % world).to(torch.int64)
        # size exchange first (overlaps the sort); one host readback total
        send_counts = torch.bincount(owner, minlength=world)
        all_counts = [torch.zeros_like(send_counts) for _ in range(world)]
        dist.all_gather(all_counts, send_counts)
        perm = torch.argsort(owner, stable=True)
        ids_sorted = g[perm]
        cnt = torch.stack(all_counts).cpu()
        in_splits = [int(c) for c in cnt[rank]]
        out_splits = [int(cnt[src][rank]) for src in range(world)]
        m = sum(out_splits)

        recv_ids = torch.empty(m, dtype=g.dtype, device=g.device)
        _all_to_all_single(recv_ids, ids_sorted.contiguous(), out_splits, in_splits)
        local_rows = torch.div(recv_ids, world, rounding_mode="floor")

        if arena1.dtype == torch.bfloat16 and use_hip(arena1):
            v1 = hip_ops().embedding_gather(arena1, local_rows.reshape(-1, 1))
            v2 = hip_ops().embedding_gather(arena2, local_rows.reshape(-1, 1))
        else:
            v1 = arena1.index_select(0, local_rows)
            v2 = arena2.index_select(0, local_rows)
        vals = torch.cat([v1.reshape(m, D1), v2.reshape(m, D2)], dim=1)

        back = torch.empty(n, D1 + D2, dtype=vals.dtype, device=vals.device)
        _all_to_all_single(back, vals.contiguous(), in_splits, out_splits)
        out = torch.empty_like(back)
        out[perm] = back

        ctx.save_for_backward(perm, local_rows)
        ctx.splits = (in_splits, out_splits)
        ctx.dims = (D1, D2)
        ctx.world = world
        ctx.shapes = (arena1.shape, arena2.shape)
        ctx.dtypes = (arena1.dtype, arena2.dtype)
        B = n // F
        return (out[:, :D1].reshape(B, F * D1).contiguous(),
                out[:, D1:].reshape(B, F * D2).contiguous())

    @staticmethod
    def backward(ctx, dout1, dout2):
        perm, local_rows = ctx.saved_tensors
        in_splits, out_splits = ctx.splits
        D1, D2 = ctx.dims
        world = ctx.world
        n = perm.numel()
        dvals = torch.cat([dout1.reshape(n, D1), dout2.reshape(n, D2)], dim=1)
        dsorted = dvals[perm].contiguous()
        grad_rows = torch.empty(local_rows.numel(), D1 + D2, dtype=dsorted.dtype,
                                device=dsorted.device)
        _all_to_all_single(grad_rows, dsorted, out_splits, in_splits)
        grad_rows = grad_rows / world
        idx = local_rows.reshape(1, -1)
        g1 = torch.sparse_coo_tensor(idx, grad_rows[:, :D1].contiguous()
                                     .to(ctx.dtypes[0]), ctx.shapes[0])
        g2 = torch.sparse_coo_tensor(idx, grad_rows[:, D1:].contiguous()
                                     .to(ctx.dtypes[1]), ctx.shapes[1])
        return g1, g2, None, None, None, None, None, None


def ep_pair_gather(emb1: "ShardedEmbedding", emb2: "ShardedEmbedding",
                   ids: torch.Tensor):
    """Shared-routing lookup for two ShardedEmbeddings over the same vocab.
    Returns ([B, F*D1], [B, F*D2])."""
    local = ids.clamp(min=0) % emb1.sizes
    flat = (local + emb1.offsets).reshape(-1)
    return _EPGatherPairFn.apply(emb1.arena, emb2.arena, flat,
                                 emb1.num_features, emb1.dim, emb2.dim,
                                 emb1.world, emb1.rank)


class ShardedEmbedding(torch.nn.Module):
    """MultiEmbedding-compatible module whose arena is sharded row%world.

    Initialization draws the SAME global arena as MultiEmbedding(seed) and
    keeps rows global_row % world == rank, so a sharded model is numerically
    identical to the replicated one."""

    def __init__(self, vocab_sizes: Sequence[int], dim: int, seed: int = 0,
                 world: int = 1, rank: int = 0,
                 dtype: torch.dtype = torch.float32, fast_init: bool = False):
        super().__init__()
        self.vocab_sizes = [int(v) for v in vocab_sizes]
        self.dim = int(dim)
        self.world, self.rank = int(world), int(rank)
        self.total_rows = int(sum(self.vocab_sizes))
        offsets = torch.tensor(
            [0] + list(torch.cumsum(torch.tensor(self.vocab_sizes), 0)[:-1]),
            dtype=torch.int64)
        self.register_buffer("offsets", offsets)
        self.register_buffer("sizes", torch.tensor(self.vocab_sizes, dtype=torch.int64))

        scale = 1.0 / math.sqrt(max(self.dim, 1))
        shard_rows = len(range(self.rank, self.total_rows, self.world))
        if fast_init:
            # per-shard seeded draw (NOT bit-identical to MultiEmbedding's
            # stream — same distribution; use for benchmarks where multi-GB
            # replicated CPU generation per rank would dominate startup)
            gen = torch.Generator().manual_seed(seed * 1000003 + rank)
            shard = (torch.rand(shard_rows, self.dim, generator=gen) * 2 - 1) * scale
        else:
            # chunked generation: identical RNG stream to MultiEmbedding's
            # single torch.rand(total, dim) call, but never materializes the
            # full arena (8 ranks x multi-GB would blow host RAM on one node)
            gen = torch.Generator().manual_seed(seed)
            shard = torch.empty(shard_rows, self.dim)
            CH = 1 << 20
            out_off = 0
            for start in range(0, self.total_rows, CH):
                n = min(CH, self.total_rows - start)
                chunk = (torch.rand(n, self.dim, generator=gen) * 2 - 1) * scale
                first = (self.rank - start) % self.world
                sel = chunk[first::self.world]
                shard[out_off:out_off + sel.shape[0]] = sel
                out_off += sel.shape[0]
        self.arena = torch.nn.Parameter(shard.to(dtype))
        self.arena._is_embedding_arena = True
        self.arena._is_ep_sharded = True   # GradAggregator: no cross-rank sync

    @property
    def num_features(self) -> int:
        return len(self.vocab_sizes)

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        if ids.shape[1] != self.num_features:
            raise ValueError(f"ids has {ids.shape[1]} features, expected {self.num_features}")
        local = ids.clamp(min=0) % self.sizes
        flat = (local + self.offsets).reshape(-1)
        return _EPGatherFn.apply(self.arena, flat, self.num_features, self.dim,
                                 self.world, self.rank)
