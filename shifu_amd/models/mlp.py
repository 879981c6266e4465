"""ModelConfig-driven MLP binary classifier.

Reproduces the reference's generate_from_modelconf + model()
(reference: ssgd_monitor.py:91-144): N hidden FusedLinear layers with
per-layer activations from ModelConfig['train']['params'], then a 1-unit
head named `shifu_output_0` (ssgd_monitor.py:121).  The head emits LOGITS;
sigmoid is fused into the loss kernel (ops/loss.py) and applied explicitly
at inference (predict_proba), so exported scores match the reference's
sigmoid output.
"""
from __future__ import annotations

from typing import List, Optional

import torch

from shifu_amd.config.model_config import ModelConfig
from shifu_amd.ops.linear import FusedLinear
from shifu_amd.ops.loss import predict_proba


class ShifuMLP(torch.nn.Module):
    def __init__(self, num_features: int, hidden_nodes: List[int],
                 activations: List[str], seed: int = 1234):
        super().__init__()
        if len(hidden_nodes) != len(activations):
            raise ValueError("hidden_nodes and activations must align")
        self.num_features = num_features
        layers = []
        prev = num_features
        for i, (h, a) in enumerate(zip(hidden_nodes, activations)):
            layers.append(FusedLinear(prev, h, activation=a, seed=seed + i))
            prev = h
        self.hidden = torch.nn.ModuleList(layers)
        # output head: 1 unit, logits (sigmoid fused in loss / applied at eval)
        self.shifu_output_0 = FusedLinear(prev, 1, activation="none",
                                          seed=seed + len(hidden_nodes))

    @classmethod
    def from_model_config(cls, mc: ModelConfig, num_features: int,
                          seed: int = 1234) -> "ShifuMLP":
        return cls(num_features, mc.params.num_hidden_nodes,
                   mc.params.activation_funcs, seed=seed)

    def forward(self, dense: torch.Tensor, cats: Optional[torch.Tensor] = None) -> torch.Tensor:
        x = dense
        for layer in self.hidden:
            x = layer(x)
        return self.shifu_output_0(x).reshape(-1)  # logits [B]

    @torch.no_grad()
    def predict(self, dense: torch.Tensor, cats: Optional[torch.Tensor] = None) -> torch.Tensor:
        return predict_proba(self.forward(dense, cats))


def build_model(mc: ModelConfig, num_dense: int, vocab_sizes=None,
                model_type: str = "mlp", embed_dim: int = 16, seed: int = 1234,
                sharded_embeddings: bool = False, world: int = 1, rank: int = 0,
                unified: bool = False):
    """Model factory over the supported families.  sharded_embeddings:
    False | "table" (feature-sharded EP) | "row" (row%world EP); unified=True
    uses the single [R, D+2] wide+deep arena (models/wide_deep.py)."""
    model_type = model_type.lower()
    vocab_sizes = list(vocab_sizes or [])
    if model_type == "mlp" or not vocab_sizes:
        return ShifuMLP.from_model_config(mc, num_dense, seed=seed)
    kw = dict(sharded_embeddings=sharded_embeddings, world=world, rank=rank,
              unified=unified)
    if model_type in ("wide_deep", "widedeep", "wnd"):
        from shifu_amd.models.wide_deep import WideDeep
        return WideDeep(num_dense, vocab_sizes, embed_dim,
                        mc.params.num_hidden_nodes, mc.params.activation_funcs,
                        seed=seed, **kw)
    if model_type == "deepfm":
        from shifu_amd.models.deepfm import DeepFM
        return DeepFM(num_dense, vocab_sizes, embed_dim,
                      mc.params.num_hidden_nodes, mc.params.activation_funcs,
                      seed=seed, **kw)
    raise ValueError(f"unknown model_type {model_type!r}")
