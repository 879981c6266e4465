"""Wide&Deep tabular binary classifier (BASELINE.json config 3 — the
headline benchmark model: 1M-vocab categorical embeddings + 200 dense).

wide  : per-category scalar weights (a D=1 embedding arena) + a linear term
        over the dense features;
deep  : concat(dense, per-feature D-dim embeddings) -> MLP tower
        (FusedLinear MFMA kernels);
head  : logits = wide + deep_head; sigmoid fused in the loss.

Not present in the reference (dense-only MLPs) — added per BASELINE.json
configs 3/4 (SURVEY.md §2.4 row "new configs only").
"""
from __future__ import annotations

from typing import List, Sequence

import torch

from shifu_amd.ops.embedding import MultiEmbedding
from shifu_amd.ops.linear import FusedLinear
from shifu_amd.ops.loss import predict_proba


def _emb_factory(vocab_sizes, sharded, world: int, rank: int, fast_init: bool):
    """sharded: False -> replicated MultiEmbedding; "table"/True -> feature-
    sharded TableShardedEmbedding (static all-to-all splits, the default EP
    mode); "row" -> row%world ShardedEmbedding (for a single table too big
    for one GPU)."""
    if not sharded or world <= 1:
        return lambda d, s: MultiEmbedding(vocab_sizes, d, seed=s)
    mode = "table" if sharded is True else str(sharded)
    if mode == "table":
        from shifu_amd.parallel.ep import TableShardedEmbedding
        return lambda d, s: TableShardedEmbedding(vocab_sizes, d, seed=s,
                                                  world=world, rank=rank,
                                                  fast_init=fast_init)
    if mode == "row":
        from shifu_amd.parallel.ep import ShardedEmbedding
        return lambda d, s: ShardedEmbedding(vocab_sizes, d, seed=s,
                                             world=world, rank=rank,
                                             fast_init=fast_init)
    raise ValueError(f"unknown sharded_embeddings mode {sharded!r}")


def make_unified_embedding(vocab_sizes, embed_dim: int, seed: int,
                           sharded, world: int, rank: int, fast_init: bool):
    """One [R, D+4] arena for wide+deep+rowwise-adagrad-accumulator
    (ops/embedding.py UnifiedMultiEmbedding docstring).  Sharded variants
    reuse the EP modules with dim=D+4 and the same per-column init scale,
    so a sharded model is numerically identical to the replicated one."""
    import math as _math
    import torch as _torch
    from shifu_amd.ops.embedding import UnifiedMultiEmbedding
    if not sharded or world <= 1:
        return UnifiedMultiEmbedding(vocab_sizes, embed_dim, seed=seed)
    mode = "table" if sharded is True else str(sharded)
    cs = _torch.ones(embed_dim + 4)
    cs[:embed_dim] = 1.0 / _math.sqrt(max(embed_dim, 1))
    cs[embed_dim + 1:] = 0.0      # pad + accumulator start at zero
    if mode == "table":
        from shifu_amd.parallel.ep import TableShardedEmbedding
        m = TableShardedEmbedding(vocab_sizes, embed_dim + 4, seed=seed,
                                  world=world, rank=rank, fast_init=fast_init,
                                  col_scale=cs)
    elif mode == "row":
        from shifu_amd.parallel.ep import ShardedEmbedding
        m = ShardedEmbedding(vocab_sizes, embed_dim + 4, seed=seed,
                             world=world, rank=rank, fast_init=fast_init,
                             col_scale=cs)
    else:
        raise ValueError(f"unknown sharded_embeddings mode {sharded!r}")
    m.arena._unified_split = embed_dim
    m.arena._acc_in_arena = True
    return m


class WideDeep(torch.nn.Module):
    """unified=True (the shipping default for training): wide scalar weights
    live as column D of the deep embedding arena — one gather / one
    scatter+adagrad chain / one EP exchange for both parts.  Exports always
    keep the LOGICAL wide/deep split (train/export.py), so serving artifacts
    are identical either way.  unified=False keeps the round-1 two-arena
    layout (and is what exported models reload as)."""

    def __init__(self, num_dense: int, vocab_sizes: Sequence[int], embed_dim: int,
                 hidden_nodes: List[int], activations: List[str], seed: int = 1234,
                 sharded_embeddings: bool = False, world: int = 1, rank: int = 0,
                 emb_fast_init: bool = False, unified: bool = False):
        super().__init__()
        self.num_dense = num_dense
        self.embed_dim = embed_dim
        self.vocab_sizes = list(vocab_sizes)
        self.unified = bool(unified)
        F = len(self.vocab_sizes)

        self.wide_dense = FusedLinear(num_dense, 1, activation="none", seed=seed + 102)
        if self.unified:
            self.embeddings = make_unified_embedding(
                self.vocab_sizes, embed_dim, seed + 201, sharded_embeddings,
                world, rank, emb_fast_init)
        else:
            emb = _emb_factory(self.vocab_sizes, sharded_embeddings, world, rank,
                               emb_fast_init)
            self.wide_cat = emb(1, seed + 101)
            self.embeddings = emb(embed_dim, seed + 201)

        tower_in = num_dense + F * embed_dim
        layers, prev = [], tower_in
        for i, (h, a) in enumerate(zip(hidden_nodes, activations)):
            layers.append(FusedLinear(prev, h, activation=a, seed=seed + 301 + i))
            prev = h
        self.tower = torch.nn.ModuleList(layers)
        self.shifu_output_0 = FusedLinear(prev, 1, activation="none", seed=seed + 999)

    def _unified_inputs(self, dense, cats):
        """(tower_in, wide_e [B,F]) through the unified arena."""
        from shifu_amd.ops.embedding import UnifiedMultiEmbedding
        D = self.embed_dim
        if isinstance(self.embeddings, UnifiedMultiEmbedding):
            return self.embeddings.gather_split(cats, dense)
        # EP-sharded unified arena: one routing pass moves [.., D+4] rows
        B = dense.shape[0]
        DP = self.embeddings.dim                     # EP module arena width
        out = self.embeddings(cats)                  # [B, F*DP]
        v = out.view(B, -1, DP)
        deep = v[:, :, :D].reshape(B, -1)
        wide_e = v[:, :, D]
        return torch.cat([dense, deep.to(dense.dtype)], dim=1), wide_e

    def forward(self, dense: torch.Tensor, cats: torch.Tensor) -> torch.Tensor:
        if self.unified:
            x, wide_e = self._unified_inputs(dense, cats)
            wide = wide_e.sum(dim=1) + self.wide_dense(dense).reshape(-1)
        else:
            from shifu_amd.parallel.ep import (ShardedEmbedding,
                                               TableShardedEmbedding,
                                               ep_pair_gather, table_pair_gather)
            if isinstance(self.wide_cat, (ShardedEmbedding, TableShardedEmbedding)):
                # shared routing: one id exchange + one combined value
                # all-to-all serves both the wide (D=1) and deep arenas
                pair = (table_pair_gather
                        if isinstance(self.wide_cat, TableShardedEmbedding)
                        else ep_pair_gather)
                wide_e, emb = pair(self.wide_cat, self.embeddings, cats)
                wide = wide_e.sum(dim=1) + self.wide_dense(dense).reshape(-1)
                x = torch.cat([dense, emb.to(dense.dtype)], dim=1)
            else:
                wide = (self.wide_cat(cats).sum(dim=1)
                        + self.wide_dense(dense).reshape(-1))
                from shifu_amd.ops.embedding import gather_concat
                x = gather_concat(self.embeddings, cats, dense)  # fused concat
        for layer in self.tower:
            x = layer(x)
        deep = self.shifu_output_0(x).reshape(-1)
        return wide.to(deep.dtype) + deep               # logits [B]

    @torch.no_grad()
    def predict(self, dense: torch.Tensor, cats: torch.Tensor) -> torch.Tensor:
        return predict_proba(self.forward(dense, cats))
