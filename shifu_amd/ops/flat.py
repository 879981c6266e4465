"""Flat parameter/gradient arena for dense parameters.

MI355X-first replacement for the reference's per-variable parameter-server
shards (variables scattered over PS tasks by replica_device_setter,
reference: ssgd_monitor.py:202-206): all dense parameters live in ONE
contiguous fp32 buffer, gradients in ONE contiguous fp32 buffer, so that

* gradient aggregation is ONE RCCL all-reduce over xGMI (or a few large
  buckets) instead of per-variable gRPC messages, and
* the optimizer is ONE fused HIP kernel over the flat buffer
  (SURVEY.md §2.4 K4), not a launch per variable.

Embedding arenas (sparse grads) are excluded — they take the sparse path in
FusedOptimizer.
"""
from __future__ import annotations

from typing import Iterable, List, Tuple

import torch


def split_params(module: torch.nn.Module) -> Tuple[List[torch.nn.Parameter],
                                                   List[torch.nn.Parameter]]:
    """(dense_params, embedding_arena_params) in deterministic order."""
    dense, sparse = [], []
    for _, p in sorted(module.named_parameters(), key=lambda kv: kv[0]):
        if not p.requires_grad:
            continue
        (sparse if getattr(p, "_is_embedding_arena", False) else dense).append(p)
    return dense, sparse


class FlatParams:
    """Flattens a list of same-dtype parameters into one buffer and rebinds
    each parameter's .data (and .grad) to a view of it."""

    def __init__(self, params: List[torch.nn.Parameter], mirror_bf16: bool = False):
        self.params = list(params)
        self.mirror = None
        if not self.params:
            self.flat = torch.zeros(0)
            self.flat_grad = torch.zeros(0)
            self._offsets: List[Tuple[int, int]] = []
            return
        dev = self.params[0].device
        dtype = self.params[0].dtype
        total = sum(p.numel() for p in self.params)
        self.flat = torch.empty(total, device=dev, dtype=dtype)
        self.flat_grad = torch.zeros(total, device=dev, dtype=dtype)
        self._offsets = []
        off = 0
        for p in self.params:
            n = p.numel()
            self.flat[off:off + n].copy_(p.data.reshape(-1))
            p.data = self.flat[off:off + n].view_as(p.data)
            # Pre-bind .grad to the flat view; AccumulateGrad then adds in
            # place.  _sync_grads() verifies the binding survived (autograd
            # may replace .grad out-of-place in edge cases) and repairs it.
            p.grad = self.flat_grad[off:off + n].view_as(p.data)
            self._offsets.append((off, n))
            off += n
        if mirror_bf16:
            # ONE bf16 compute copy of the whole arena, refreshed by a single
            # cast kernel per step (vs a small cast kernel per layer) — the
            # fused-linear forward reads per-param views of it.
            self.mirror = self.flat.to(torch.bfloat16)

    def refresh_mirror(self) -> None:
        if self.mirror is not None:
            self.mirror.copy_(self.flat)

    def mirror_view(self, p: torch.nn.Parameter):
        if self.mirror is None:
            return None
        i = next(i for i, q in enumerate(self.params) if q is p)
        off, n = self._offsets[i]
        return self.mirror[off:off + n].view_as(p.data)

    def grad_view(self, p: torch.nn.Parameter):
        i = next(i for i, q in enumerate(self.params) if q is p)
        off, n = self._offsets[i]
        return self.flat_grad[off:off + n].view_as(p.data)

    def numel(self) -> int:
        return self.flat.numel()

    def zero_grad(self) -> None:
        if self.flat_grad.numel():
            self.flat_grad.zero_()
        for p, (off, n) in zip(self.params, self._offsets):
            if p.grad is None or p.grad.data_ptr() != self.flat_grad[off:off + n].data_ptr():
                p.grad = self.flat_grad[off:off + n].view_as(p.data)

    def sync_grads(self) -> None:
        """Ensure flat_grad holds the accumulated grads (repair any view that
        autograd replaced out-of-place)."""
        for p, (off, n) in zip(self.params, self._offsets):
            if p.grad is None:
                continue
            view = self.flat_grad[off:off + n]
            if p.grad.data_ptr() != view.data_ptr():
                view.copy_(p.grad.reshape(-1))
                p.grad = view.view_as(p.data)

    def grad_views(self) -> Iterable[torch.Tensor]:
        for off, n in self._offsets:
            yield self.flat_grad[off:off + n]


def bind_mirrors(module: torch.nn.Module, flat: "FlatParams") -> None:
    """Attach per-layer bf16 mirror views to every FusedLinear whose params
    live in this arena (no-op without a mirror)."""
    if flat.mirror is None:
        return
    from shifu_amd.ops.linear import FusedLinear
    pset = {id(p): i for i, p in enumerate(flat.params)}
    for m in module.modules():
        if isinstance(m, FusedLinear) and id(m.weight) in pset and id(m.bias) in pset:
            m._w_mirror = flat.mirror_view(m.weight)
            m._b_mirror = flat.mirror_view(m.bias)
            if m.out_features > 1:  # heads keep the autograd GEMV path
                m._w_gradview = flat.grad_view(m.weight)
                m._b_gradview = flat.grad_view(m.bias)
