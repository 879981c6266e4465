"""Categorical embedding arena: gather forward, sparse rowwise gradients.

New-config op (Wide&Deep / DeepFM — BASELINE.json configs 3/4; the reference
is dense-only, SURVEY.md §2.4 row "new configs only").

Design (MI355X-first):
* ALL F categorical tables live in ONE arena tensor [sum(V_f), D]
  (bf16 on GPU — a 26x1M x 64 arena is 3.3 GB of the 288 GB HBM; fp32 on CPU),
  with per-feature row offsets.  One gather kernel serves every feature.
* forward: ids [B, F] -> out [B, F*D] (concatenated), one coalesced
  gather kernel (D*2B contiguous bytes per row).
* backward: returns a torch SPARSE grad (global row indices + dout rows) —
  a dense grad would be the full arena (GBs) and its all-reduce would
  dominate the step.  The FusedOptimizer aggregates sparse grads across
  ranks (allgather of rows over xGMI) and applies a rowwise update kernel
  (sort-free packed-bf16 atomic scatter on GPU — SURVEY.md §2.4 embedding row).
"""
from __future__ import annotations

import math
from typing import Sequence

import torch

from shifu_amd.ops.dispatch import use_hip, hip_ops


def sparse_rows_values(g: torch.Tensor):
    """(rows int64, values [n,D] in grad dtype) from a sparse grad.

    Rows MAY CONTAIN DUPLICATES: every consumer tolerates them — the HIP
    update kernels scatter with atomics (global_atomic_pk_add_bf16 / CAS),
    the CPU path uses index_add_, and updates are linear in the gradient.
    Skipping the dedup avoids a rocprim sort (torch.unique) per step, which
    profiling showed at ~20% of the Wide&Deep step.  Values keep their
    native dtype (bf16 on GPU): the update kernels read bf16 directly and
    the DP allgather wires half the bytes."""
    if not g.is_sparse:
        rows = torch.nonzero(g.abs().sum(dim=1) != 0, as_tuple=False).reshape(-1)
        return rows, g[rows]
    return g._indices()[0], g._values()


class _EmbGatherFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, arena: torch.Tensor, flat_ids: torch.Tensor, F: int, D: int):
        # flat_ids: [B, F] global row ids (offsets already added)
        B = flat_ids.shape[0]
        if arena.dtype == torch.bfloat16 and use_hip(arena):
            ext = hip_ops()
            out = ext.embedding_gather(arena, flat_ids.contiguous())
        else:
            # fp32 arena (explicit rc.dtype="fp32" config) takes the eager path
            out = arena.index_select(0, flat_ids.reshape(-1)).reshape(B, F * D)
        ctx.save_for_backward(flat_ids)
        ctx.arena_shape = arena.shape
        ctx.F, ctx.D = F, D
        return out

    @staticmethod
    def backward(ctx, dout: torch.Tensor):
        (flat_ids,) = ctx.saved_tensors
        B = flat_ids.shape[0]
        values = dout.reshape(B * ctx.F, ctx.D)
        indices = flat_ids.reshape(1, B * ctx.F)
        grad = torch.sparse_coo_tensor(indices, values, ctx.arena_shape)
        return grad, None, None, None


class _EmbGatherConcatFn(torch.autograd.Function):
    """Fused [dense | embeddings] tower-input build: the gather kernel writes
    straight into the concat buffer (a separate torch.cat pass over the
    [B, nd+F*D] activations measured ~4% of the Wide&Deep step)."""

    @staticmethod
    def forward(ctx, arena, flat_ids, dense, F, D):
        B, nd = dense.shape
        ext = hip_ops()
        out = torch.empty(B, nd + F * D, device=dense.device, dtype=dense.dtype)
        out[:, :nd].copy_(dense)
        ext.embedding_gather_into(arena, flat_ids.contiguous(), out, nd)
        ctx.save_for_backward(flat_ids)
        ctx.arena_shape = arena.shape
        ctx.F, ctx.D, ctx.nd = F, D, nd
        return out

    @staticmethod
    def backward(ctx, dout):
        (flat_ids,) = ctx.saved_tensors
        B = flat_ids.shape[0]
        values = dout[:, ctx.nd:].reshape(B * ctx.F, ctx.D)
        grad = torch.sparse_coo_tensor(flat_ids.reshape(1, B * ctx.F),
                                       values, ctx.arena_shape)
        # dense input carries no grad in training (raw features)
        return grad, None, None, None, None


def gather_concat(emb: "MultiEmbedding", ids: torch.Tensor,
                  dense: torch.Tensor) -> torch.Tensor:
    """[dense | emb(ids)] in one pass when the fused kernel applies;
    falls back to gather + torch.cat."""
    if (emb.arena.dtype == torch.bfloat16 and dense.dtype == torch.bfloat16
            and emb.dim % 8 == 0 and use_hip(emb.arena)
            and not dense.requires_grad):
        local = ids.clamp(min=0) % emb.sizes
        flat = local + emb.offsets
        return _EmbGatherConcatFn.apply(emb.arena, flat, dense.contiguous(),
                                        emb.num_features, emb.dim)
    out = emb(ids)
    return torch.cat([dense, out.to(dense.dtype)], dim=1)


class _UnifiedGatherFn(torch.autograd.Function):
    """ONE [R, D+2] arena (deep cols 0..D-1, wide col D, zero pad D+1) ->
    (tower_in [B, nd+F*D], wide [B, F]) in a single gather pass
    (emb_gather_split kernel).  Backward packs the deep and wide output
    grads into one sparse [n, D+2] value tensor, so the whole wide+deep
    update runs ONE scatter/adagrad chain instead of round 1's two."""

    @staticmethod
    def forward(ctx, arena: torch.Tensor, flat_ids: torch.Tensor,
                dense: torch.Tensor, F: int, D: int, defer: bool = False):
        B, nd = dense.shape
        hip = (arena.dtype == torch.bfloat16 and use_hip(arena)
               and D % 8 == 0 and dense.dtype == torch.bfloat16
               and not dense.requires_grad)
        if hip:
            ext = hip_ops()
            out = torch.empty(B, nd + F * D, device=dense.device,
                              dtype=dense.dtype)
            out[:, :nd].copy_(dense)
            wide = torch.empty(B, F, device=dense.device, dtype=arena.dtype)
            ext.emb_gather_split(arena, flat_ids.contiguous(), out, wide,
                                 nd, D)
        else:
            g = arena.index_select(0, flat_ids.reshape(-1))      # [n, DP]
            deep = g[:, :D].reshape(B, F * D)
            wide = g[:, D].reshape(B, F)
            out = torch.cat([dense, deep.to(dense.dtype)], dim=1)
        ctx.save_for_backward(flat_ids)
        ctx.meta = (arena.shape, arena.dtype, F, D, nd)
        ctx.defer = bool(defer) and hip
        ctx.arena_ref = arena if ctx.defer else None
        return out, wide

    @staticmethod
    def backward(ctx, dout: torch.Tensor, dwide: torch.Tensor):
        (flat_ids,) = ctx.saved_tensors
        shape, dtype, F, D, nd = ctx.meta
        B = dout.shape[0]
        n = B * F
        DP = shape[1]
        if ctx.defer:
            # fast path (single-rank, no window accumulation): hand the
            # UNPACKED grad buffers straight to the optimizer's unified
            # update kernels — no [n, D+2] value materialization, no 132B
            # row stride in the scatter (ops/hip emb_update_unified)
            lst = getattr(ctx.arena_ref, "_unified_grads", None)
            if lst is None:
                lst = []
                ctx.arena_ref._unified_grads = lst
            lst.append((flat_ids.reshape(-1), dout, nd, dwide.contiguous(),
                        F, D))
            return None, None, None, None, None, None
        vals = torch.empty(n, DP, dtype=dtype, device=dout.device)
        v3 = vals.view(B, F, DP)
        v3[:, :, :D] = dout[:, nd:].reshape(B, F, D)
        v3[:, :, D] = dwide.to(dtype)
        v3[:, :, D + 1:] = 0
        grad = torch.sparse_coo_tensor(flat_ids.reshape(1, n), vals, shape)
        return grad, None, None, None, None, None


class UnifiedMultiEmbedding(torch.nn.Module):
    """Wide&Deep unified arena (ROADMAP item 3): per category ONE row holds
    the deep D-vector, the wide scalar weight and a zero pad column
    (4B-even rows for the packed-bf16 atomics).  One gather / one
    scatter+adagrad chain serve both parts; at world>1 this also halves the
    EP collective count vs the round-1 two-arena pair gather.

    OPTIMIZER SEMANTICS: rowwise adagrad now normalizes the wide weight by
    the row's COMBINED mean-square gradient instead of its own — not
    numerically identical to the split arenas (gate: tools/auc_parity.py).

    Exports keep the LOGICAL split (train/export.py splits the arena into
    embeddings[R,D] + wide_cat[R,1]) so serving and the Java eval bundle
    layout are unchanged."""

    def __init__(self, vocab_sizes: Sequence[int], dim: int, seed: int = 0,
                 dtype: torch.dtype = torch.float32, empty_init: bool = False):
        super().__init__()
        self.vocab_sizes = [int(v) for v in vocab_sizes]
        self.dim = int(dim)             # DEEP dim; arena has dim+4 columns:
        # [0..D) deep | D wide | D+1 zero pad | D+2..D+3 rowwise-adagrad
        # accumulator (raw f32 bits in two bf16 slots on GPU, a plain f32 in
        # col D+2 on fp32/CPU arenas).  Keeping the accumulator IN the row
        # puts the whole update chain (accsq atomic, denominator read,
        # scatter) on one cacheline neighborhood per row instead of a second
        # random-access array.
        self.cols = self.dim + 4
        self.total_rows = int(sum(self.vocab_sizes))
        offsets = torch.tensor(
            [0] + list(torch.cumsum(torch.tensor(self.vocab_sizes), 0)[:-1]),
            dtype=torch.int64)
        self.register_buffer("offsets", offsets, persistent=False)
        self.register_buffer("sizes", torch.tensor(self.vocab_sizes, dtype=torch.int64),
                             persistent=False)
        if empty_init:
            arena = torch.empty(self.total_rows, self.cols)
        else:
            gen = torch.Generator().manual_seed(seed)
            arena = (torch.rand(self.total_rows, self.cols, generator=gen) * 2 - 1)
            s = torch.ones(self.cols)
            s[:self.dim] = 1.0 / math.sqrt(max(self.dim, 1))
            s[self.dim + 1:] = 0.0     # pad + accumulator start at zero
            arena *= s
        self.arena = torch.nn.Parameter(arena.to(dtype))
        self.arena._is_embedding_arena = True
        self.arena._unified_split = self.dim  # export: cols [:D] deep, [D] wide
        self.arena._acc_in_arena = True       # optimizer: no external emb_state
        # deferred-grad fast path: trainer/bench enable it when world==1 and
        # update_window==1 (the optimizer then consumes the unpacked grad
        # buffers directly — ops/hip emb_update_unified)
        self.defer_grads = False

    @property
    def num_features(self) -> int:
        return len(self.vocab_sizes)

    def flat_ids(self, ids: torch.Tensor) -> torch.Tensor:
        local = ids.clamp(min=0) % self.sizes
        return local + self.offsets

    def adagrad_acc(self) -> torch.Tensor:
        """Decode the in-row rowwise-adagrad accumulator -> [R] f32."""
        D = self.dim
        if self.arena.dtype == torch.bfloat16:
            raw = self.arena.data[:, D + 2:D + 4].contiguous()
            return raw.view(torch.float32).reshape(-1)
        return self.arena.data[:, D + 2].clone()

    def gather_split(self, ids: torch.Tensor, dense: torch.Tensor):
        """-> (tower_in [B, nd+F*D], wide [B, F])."""
        if ids.shape[1] != self.num_features:
            raise ValueError(f"ids has {ids.shape[1]} features, expected "
                             f"{self.num_features}")
        return _UnifiedGatherFn.apply(self.arena, self.flat_ids(ids), dense,
                                      self.num_features, self.dim,
                                      self.defer_grads)


def unified_col_scale(dim: int) -> torch.Tensor:
    """Per-column init scale of a unified arena: deep cols like a D-dim
    MultiEmbedding (1/sqrt(D)), wide col like the old D=1 arena (scale 1),
    pad col zero."""
    s = torch.ones(dim + 2)
    s[:dim] = 1.0 / math.sqrt(max(dim, 1))
    s[dim + 1] = 0.0
    return s


class MultiEmbedding(torch.nn.Module):
    """F categorical features -> concatenated [B, F*D] embeddings from one arena."""

    def __init__(self, vocab_sizes: Sequence[int], dim: int,
                 seed: int = 0, device: str = "cpu", dtype: torch.dtype = torch.float32,
                 empty_init: bool = False):
        super().__init__()
        self.vocab_sizes = [int(v) for v in vocab_sizes]
        self.dim = int(dim)
        self.total_rows = int(sum(self.vocab_sizes))
        offsets = torch.tensor(
            [0] + list(torch.cumsum(torch.tensor(self.vocab_sizes), 0)[:-1]),
            dtype=torch.int64)
        # topology buffers are derived from vocab_sizes — never checkpointed
        # (persistent=False keeps state_dicts arena-only, so replicated and
        # sharded layouts interconvert cleanly)
        self.register_buffer("offsets", offsets, persistent=False)
        # device-resident so forward makes no host->device copies (hipGraph-safe)
        self.register_buffer("sizes", torch.tensor(self.vocab_sizes, dtype=torch.int64),
                             persistent=False)
        if empty_init:
            # caller overwrites (e.g. EP export consolidation) — skip the
            # multi-GB random draw
            arena = torch.empty(self.total_rows, self.dim)
        else:
            gen = torch.Generator().manual_seed(seed)
            scale = 1.0 / math.sqrt(max(self.dim, 1))
            arena = (torch.rand(self.total_rows, self.dim, generator=gen) * 2 - 1) * scale
        self.arena = torch.nn.Parameter(arena.to(dtype))
        self.arena._is_embedding_arena = True  # FusedOptimizer routes this to the sparse path

    @property
    def num_features(self) -> int:
        return len(self.vocab_sizes)

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        # ids: [B, F] per-feature local ids; clamp out-of-vocab to last row
        if ids.shape[1] != self.num_features:
            raise ValueError(f"ids has {ids.shape[1]} features, expected {self.num_features}")
        local = ids.clamp(min=0) % self.sizes  # hash-style fold of out-of-range ids
        flat = local + self.offsets
        return _EmbGatherFn.apply(self.arena, flat, self.num_features, self.dim)
