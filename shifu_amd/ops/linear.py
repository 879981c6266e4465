"""Fused dense layer: Y = act(X @ W^T + b), weights stored [out, in].

Replaces the reference's per-layer `tf.matmul(x,W)+b` + activation
(reference: ssgd_monitor.py:57-71 nn_layer, activations ssgd_monitor.py:74-88)
with hand-written CDNA4 kernels.

Layout strategy (MI355X-first): every hot GEMM runs the v3 "NT" kernel whose
two operands are both stored reduction-major (global_load_lds staging, no
scatter-transposes):
  fwd   : y  = nt(x [B,K], w [N,K]) + bias/act epilogue
  dgrad : dx = nt(dz [B,N], w^T [K,N])        (w^T = one small per-step
                                               tiled-transpose kernel)
  wgrad : dw = nt(dz^T [N,B], x^T [K,B])      (activation transposes, f32
                                               out, split-K; the transpose-
                                               free ttv3 alternative measured
                                               slower — see _WGRAD_TT below)
Mixed precision: fp32 master weights (autograd leaves), bf16 compute via
per-layer views of the flat arena's bf16 mirror (ONE arena-wide cast per
step — ops/flat.py), fp32 dW/db, bf16 dX.  Weight/bias grads can run on a
side HIP stream (async wgrad) accumulated directly into the flat-grad
arena, overlapping the dgrad chain.

Activation gradients are computed from Y (not Z): sigmoid' = y(1-y),
tanh' = 1-y^2, relu'/leakyrelu' from sign(y) — valid because all four
activations are monotone with act(z)>0 <=> z>0 for (leaky)relu.
"""
from __future__ import annotations

import math
import os
from typing import Optional

import torch

from shifu_amd.ops.dispatch import use_hip, hip_ops

# ---------------------------------------------------------------------------
# async wgrad: weight/bias gradients are computed on a side HIP stream and
# accumulated DIRECTLY into the flat-grad arena views, overlapping the
# dgrad chain (and, multi-GPU, the all-reduce) that continues on the main
# stream.  autograd then receives None for w/b (no AccumulateGrad work).
# Consumers (optimizer / aggregator) call drain_wgrad_events() before
# touching the flat grads.
# ---------------------------------------------------------------------------
# OFF by default: measured -30% on deep dense towers (the side stream
# becomes the critical path and competes for CUs) and ~neutral on Wide&Deep;
# with it off, per-bucket all-reduce/backward overlap via the accumulate-grad
# hooks applies on multi-GPU.  Opt in with SHIFU_ASYNC_WGRAD=1.
_ASYNC_WGRAD = os.environ.get("SHIFU_ASYNC_WGRAD", "0") == "1"

# Transpose-free wgrad (ttv3): dw += dz^T @ x with the batch-major operands
# scatter-staged into the same swizzled LDS image the NT kernels use — no
# xT/dzT copies or transpose launches.  MEASURED SLOWER than the transpose
# route and kept OFF: 116 vs 108 us at [1024,1864,red 8192] and 431 vs 257 us
# at red 32768 (chain numbers INCLUDE both transposes); end-to-end 16.8M vs
# 20.0M samples/s.  The 256 MB L3 keeps the transposed copies LLC-resident,
# so the chain's extra traffic is nearly free, while ttv3's b16 scatter
# staging gives up the glds direct-to-LDS path.  SHIFU_WGRAD_TT=1 enables.
_WGRAD_TT = os.environ.get("SHIFU_WGRAD_TT", "0") == "1"
_WGRAD_STREAM = None
_WGRAD_EVENTS: list = []


def _wgrad_stream():
    global _WGRAD_STREAM
    if _WGRAD_STREAM is None:
        _WGRAD_STREAM = torch.cuda.Stream()
    return _WGRAD_STREAM


def drain_wgrad_events(stream=None) -> None:
    """Make `stream` (default: current) wait for all pending async wgrads."""
    if not _WGRAD_EVENTS:
        return
    s = stream or torch.cuda.current_stream()
    for ev in _WGRAD_EVENTS:
        s.wait_event(ev)
    _WGRAD_EVENTS.clear()

# activation ids shared with the HIP side (ops/hip/shifu_ops.hip)
ACT_NONE, ACT_SIGMOID, ACT_TANH, ACT_RELU, ACT_LEAKYRELU = 0, 1, 2, 3, 4
_ACT_IDS = {"none": ACT_NONE, "linear": ACT_NONE, "sigmoid": ACT_SIGMOID,
            "tanh": ACT_TANH, "relu": ACT_RELU, "leakyrelu": ACT_LEAKYRELU}
LEAKY_SLOPE = 0.01  # tf.nn.leaky_relu default alpha (ssgd_monitor.py:86)


def act_id(name: str) -> int:
    return _ACT_IDS[name.lower()]


def _dp_overlap_active() -> bool:
    """True when multi-rank bucketed all-reduce overlap is live: wgrad must
    then flow through AccumulateGrad so the per-bucket hooks fire."""
    import torch.distributed as dist
    return dist.is_available() and dist.is_initialized() and dist.get_world_size() > 1


def _act_fwd_ref(z: torch.Tensor, act: int) -> torch.Tensor:
    if act == ACT_SIGMOID:
        return torch.sigmoid(z)
    if act == ACT_TANH:
        return torch.tanh(z)
    if act == ACT_RELU:
        return torch.relu(z)
    if act == ACT_LEAKYRELU:
        return torch.nn.functional.leaky_relu(z, LEAKY_SLOPE)
    return z


def _act_grad_from_y_ref(dy: torch.Tensor, y: torch.Tensor, act: int) -> torch.Tensor:
    if act == ACT_SIGMOID:
        return dy * y * (1.0 - y)
    if act == ACT_TANH:
        return dy * (1.0 - y * y)
    if act == ACT_RELU:
        return dy * (y > 0).to(dy.dtype)
    if act == ACT_LEAKYRELU:
        return dy * torch.where(y > 0, torch.ones_like(y),
                                torch.full_like(y, LEAKY_SLOPE))
    return dy


class _FusedLinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, w: torch.Tensor, b: torch.Tensor, act: int,
                w_mirror=None, b_mirror=None, w_gradview=None, b_gradview=None):
        # w: [out, in].  w_mirror/b_mirror: bf16 views of the flat arena's
        # compute copy (refreshed once per step by the optimizer) — when
        # absent, cast per call.
        if use_hip(x):
            ext = hip_ops()
            wb = (w_mirror if w_mirror is not None else w.to(torch.bfloat16)).contiguous()
            bb = (b_mirror if b_mirror is not None else b.to(torch.bfloat16)).contiguous()
            xb = x if x.dtype == torch.bfloat16 else x.to(torch.bfloat16)
            xb = xb.contiguous()
            if w.shape[0] == 1:   # 1-unit head: GEMV path (memory-speed)
                y = ext.gemv_fwd(xb, wb, bb, act)
            else:
                y = ext.linear_nt_fwd(xb, wb, bb, act)
            ctx.save_for_backward(xb, wb, y)
        else:
            z = x @ w.t() + b
            y = _act_fwd_ref(z, act)
            ctx.save_for_backward(x, w, y)
        ctx.act = act
        ctx.hip = use_hip(x)
        ctx.x_needs_grad = x.requires_grad
        ctx.gradviews = (w_gradview, b_gradview)
        return y

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        x, w, y = ctx.saved_tensors
        act = ctx.act
        if ctx.hip:
            ext = hip_ops()
            gv_w, gv_b = ctx.gradviews
            if w.shape[0] == 1:   # 1-unit head: GEMV backward
                dz = ext.act_grad(dy.contiguous(), y, act).reshape(-1)
                dw, db, dx = ext.gemv_bwd(x, w.reshape(-1), dz, ctx.x_needs_grad)
                if not ctx.x_needs_grad:
                    dx = None
                return dx, dw, db, None, None, None, None, None
            # transpose-free wgrad when both weight dims are 16 B-aligned:
            # dw comes from the ttv3 kernel on the batch-major dz/x directly
            tt = _WGRAD_TT and w.shape[0] % 8 == 0 and w.shape[1] % 8 == 0
            # one fused pass: dz = dy*act'(y) (row-major AND, on the
            # transpose route, transposed) + db = colsum(dz)
            if (gv_w is not None and gv_b is not None and not _ASYNC_WGRAD
                    and not _dp_overlap_active()):
                # single-rank fast path: colsum accumulates into the bias
                # flat-grad view and the split-K wgrad reduce accumulates
                # into the weight view — autograd gets None for w/b (no
                # fresh dw alloc, no AccumulateGrad adds, no db zeros)
                if tt:
                    dz = ext.act_grad_colsum_into(dy.contiguous(), y, act,
                                                  gv_b)
                    ext.gemm_ttv3_f32_into(dz, x, gv_w)
                else:
                    dz, dzT = ext.act_grad_colsum_T_into(dy.contiguous(), y,
                                                         act, gv_b)
                    xT = ext.transpose_bf16(x)
                    ext.gemm_ntv3_f32_into(dzT, xT, gv_w)
                dx = None
                if ctx.x_needs_grad:
                    wT = ext.transpose_bf16(w)
                    dx = ext.gemm_ntv3_bf16(dz, wT)
                return dx, None, None, None, None, None, None, None
            if tt:
                dz, db = ext.act_grad_colsum(dy.contiguous(), y, act)
                dzT = None
            else:
                dz, dzT, db = ext.act_grad_colsum_T(dy.contiguous(), y, act)
            if _ASYNC_WGRAD and gv_w is not None and gv_b is not None:
                # wgrad on a side stream, accumulated straight into the flat
                # arena; autograd gets None (no AccumulateGrad for w/b)
                main = torch.cuda.current_stream()
                ev = torch.cuda.Event()
                ev.record(main)
                ws = _wgrad_stream()
                ws.wait_event(ev)
                with torch.cuda.stream(ws):
                    if tt:
                        dwv = ext.gemm_ttv3_f32(dz, x)
                    else:
                        xT = ext.transpose_bf16(x)
                        dwv = ext.gemm_ntv3_f32(dzT, xT)
                    gv_w.add_(dwv)
                    gv_b.add_(db)
                    done = torch.cuda.Event()
                    done.record(ws)
                # caching-allocator safety: main-stream tensors used on ws
                for t in (x, dz if tt else dzT, db):
                    t.record_stream(ws)
                _WGRAD_EVENTS.append(done)
                dx = None
                if ctx.x_needs_grad:
                    wT = ext.transpose_bf16(w)
                    dx = ext.gemm_ntv3_bf16(dz, wT)
                return dx, None, None, None, None, None, None, None
            if tt:
                dw = ext.gemm_ttv3_f32(dz, x)         # fp32 [N,K], split-K
            else:
                xT = ext.transpose_bf16(x)            # [K,B]
                dw = ext.gemm_ntv3_f32(dzT, xT)       # fp32 [N,K], split-K
            dx = None
            if ctx.x_needs_grad:
                wT = ext.transpose_bf16(w)            # [K,N]
                dx = ext.gemm_ntv3_bf16(dz, wT)       # bf16 [B,K]
        else:
            dz = _act_grad_from_y_ref(dy, y, act)
            dw = dz.t() @ x                           # [N,K]
            db = dz.sum(dim=0)
            dx = dz @ w if ctx.x_needs_grad else None
        return dx, dw, db, None, None, None, None, None


def fused_linear(x: torch.Tensor, w: torch.Tensor, b: torch.Tensor,
                 activation: str = "none") -> torch.Tensor:
    """y = act(x @ w^T + b) with w stored [out, in]."""
    return _FusedLinearFn.apply(x, w, b, act_id(activation))


class FusedLinear(torch.nn.Module):
    """Dense layer with xavier init + fused forward; weight stored [out, in].

    Init matches the reference's nn_layer: xavier/glorot for W
    (tf.contrib.layers.xavier_initializer, ssgd_monitor.py:63) and zero bias.
    The L2(0.1) regularizer on W (ssgd_monitor.py:58-68) is folded into the
    optimizer as coupled weight decay (grad += l2 * w).
    """

    def __init__(self, in_features: int, out_features: int,
                 activation: str = "none", seed: Optional[int] = None):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.activation = activation.lower()
        self._act = act_id(self.activation)
        gen = None
        if seed is not None:
            gen = torch.Generator().manual_seed(seed)
        limit = math.sqrt(6.0 / (in_features + out_features))
        w = (torch.rand(out_features, in_features, generator=gen) * 2 - 1) * limit
        self.weight = torch.nn.Parameter(w)                    # [N, K]
        self.bias = torch.nn.Parameter(torch.zeros(out_features))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return _FusedLinearFn.apply(x, self.weight, self.bias, self._act,
                                    getattr(self, "_w_mirror", None),
                                    getattr(self, "_b_mirror", None),
                                    getattr(self, "_w_gradview", None),
                                    getattr(self, "_b_gradview", None))

    def extra_repr(self) -> str:
        return f"in={self.in_features}, out={self.out_features}, act={self.activation}"
