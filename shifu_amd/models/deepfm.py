"""DeepFM-style tabular binary classifier (BASELINE.json config 4).

fm-1st : per-category scalar weights + linear over dense;
fm-2nd : factorization-machine pairwise interaction over the F per-feature
         D-dim embedding vectors: 0.5 * sum_d[(sum_f v_fd)^2 - sum_f v_fd^2];
deep   : same embeddings concat dense -> MLP tower;
head   : logits = fm1 + fm2 + deep_head.
"""
from __future__ import annotations

from typing import List, Sequence

import torch

from shifu_amd.ops.embedding import MultiEmbedding
from shifu_amd.ops.linear import FusedLinear
from shifu_amd.ops.loss import predict_proba


class DeepFM(torch.nn.Module):
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

        self.fm_dense = FusedLinear(num_dense, 1, activation="none", seed=seed + 12)
        if self.unified:
            # fm first-order weights = the wide column of ONE [R, D+2] arena
            from shifu_amd.models.wide_deep import make_unified_embedding
            self.embeddings = make_unified_embedding(
                self.vocab_sizes, embed_dim, seed + 21, sharded_embeddings,
                world, rank, emb_fast_init)
        else:
            from shifu_amd.models.wide_deep import _emb_factory
            emb = _emb_factory(self.vocab_sizes, sharded_embeddings, world, rank,
                               emb_fast_init)
            self.fm_first = emb(1, seed + 11)
            self.embeddings = emb(embed_dim, seed + 21)

        tower_in = num_dense + F * embed_dim
        layers, prev = [], tower_in
        for i, (h, a) in enumerate(zip(hidden_nodes, activations)):
            layers.append(FusedLinear(prev, h, activation=a, seed=seed + 31 + i))
            prev = h
        self.tower = torch.nn.ModuleList(layers)
        self.shifu_output_0 = FusedLinear(prev, 1, activation="none", seed=seed + 99)

    def forward(self, dense: torch.Tensor, cats: torch.Tensor) -> torch.Tensor:
        B = dense.shape[0]
        F = len(self.vocab_sizes)
        from shifu_amd.parallel.ep import (ShardedEmbedding,
                                           TableShardedEmbedding,
                                           ep_pair_gather, table_pair_gather)
        from shifu_amd.ops.fm import fm_second_order
        if self.unified:
            from shifu_amd.models.wide_deep import WideDeep
            x, fm1_e = WideDeep._unified_inputs(self, dense, cats)
            fm1 = fm1_e.sum(dim=1) + self.fm_dense(dense).reshape(-1)
            emb_view = x[:, self.num_dense:]                 # strided view
            fm2 = fm_second_order(emb_view, F, self.embed_dim)
            for layer in self.tower:
                x = layer(x)
            deep = self.shifu_output_0(x).reshape(-1)
            return fm1.to(deep.dtype) + fm2.to(deep.dtype) + deep
        if isinstance(self.fm_first, (ShardedEmbedding, TableShardedEmbedding)):
            pair = (table_pair_gather
                    if isinstance(self.fm_first, TableShardedEmbedding)
                    else ep_pair_gather)
            fm1_e, emb_flat = pair(self.fm_first, self.embeddings, cats)
            fm1 = fm1_e.sum(dim=1) + self.fm_dense(dense).reshape(-1)
            fm2 = fm_second_order(emb_flat, F, self.embed_dim)
            x = torch.cat([dense, emb_flat.to(dense.dtype)], dim=1)
        else:
            fm1 = self.fm_first(cats).sum(dim=1) + self.fm_dense(dense).reshape(-1)
            from shifu_amd.ops.embedding import gather_concat
            x = gather_concat(self.embeddings, cats, dense)  # [B, nd+F*D] fused
            emb_view = x[:, self.num_dense:]                 # strided view
            fm2 = fm_second_order(emb_view, F, self.embed_dim)
        for layer in self.tower:
            x = layer(x)
        deep = self.shifu_output_0(x).reshape(-1)
        return fm1.to(deep.dtype) + fm2.to(deep.dtype) + deep

    @torch.no_grad()
    def predict(self, dense: torch.Tensor, cats: torch.Tensor) -> torch.Tensor:
        return predict_proba(self.forward(dense, cats))
