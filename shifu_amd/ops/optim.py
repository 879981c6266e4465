"""Fused optimizers over the flat parameter arena + rowwise sparse embedding updates.

Dense path: ONE kernel per step over the flat fp32
master/grad/moment buffers — Adam, Adadelta (TF semantics, the reference
default: tf.train.AdadeltaOptimizer, ssgd_monitor.py:138), and SGD.  The
reference's per-layer l2_regularizer(0.1) (ssgd_monitor.py:58-68) is folded
in as coupled weight decay: g += l2 * w (tf l2_regularizer(s)(w) = s*||w||^2/2
=> d/dw = s*w).

Sparse path: embedding arenas get rowwise updates on UNCOALESCED sparse
grads (rows may repeat; the kernels scatter with packed-bf16 atomics so no
sort/dedup runs on the hot path): "sgd" or rowwise "adagrad" (fp32
accumulator per row).
"""
from __future__ import annotations

import math
from typing import Dict, List, Optional

import torch

from shifu_amd.ops.dispatch import use_hip, hip_ops
from shifu_amd.ops.flat import FlatParams

OPT_SGD, OPT_ADAM, OPT_ADADELTA, OPT_ADAGRAD = 0, 1, 2, 3
_OPT_IDS = {"sgd": OPT_SGD, "adam": OPT_ADAM, "adadelta": OPT_ADADELTA,
            "adagrad": OPT_ADAGRAD}


class FusedOptimizer:
    """Optimizer over (FlatParams dense arena, embedding arena params)."""

    def __init__(self, flat: FlatParams, emb_params: Optional[List[torch.nn.Parameter]] = None,
                 optimizer: str = "adadelta", lr: float = 1.0, l2_reg: float = 0.0,
                 betas=(0.9, 0.999), eps: float = 1e-8, rho: float = 0.95,
                 emb_optimizer: str = "adagrad", emb_lr: Optional[float] = None):
        self.flat = flat
        self.emb_params = list(emb_params or [])
        self.kind = _OPT_IDS[optimizer.lower()]
        self.lr = float(lr)
        self.l2 = float(l2_reg)
        self.b1, self.b2 = betas
        self.eps = float(eps)
        self.rho = float(rho)
        self.step_count = 0
        self.emb_kind = _OPT_IDS[emb_optimizer.lower()]
        self.emb_lr = float(emb_lr if emb_lr is not None else lr)

        n = flat.numel()
        dev = flat.flat.device if n else torch.device("cpu")
        # moment buffers (fp32): adam: m,v ; adadelta: accum, delta_accum ; adagrad: m
        self.m = torch.zeros(n, device=dev) if self.kind in (OPT_ADAM, OPT_ADADELTA, OPT_ADAGRAD) else None
        self.v = torch.zeros(n, device=dev) if self.kind in (OPT_ADAM, OPT_ADADELTA) else None
        # rowwise fp32 accumulators for embedding adagrad (arenas with the
        # D+4 unified layout carry their accumulator INSIDE the row instead)
        self.emb_state: Dict[int, torch.Tensor] = {}
        if self.emb_kind == OPT_ADAGRAD:
            for i, p in enumerate(self.emb_params):
                if not getattr(p, "_acc_in_arena", False):
                    self.emb_state[i] = torch.zeros(p.shape[0], device=p.device)

    # ------------------------------------------------------------------ dense
    def _dense_step_ref(self) -> None:
        w, g = self.flat.flat, self.flat.flat_grad
        if self.l2 != 0.0:
            g = g.add(w, alpha=self.l2)
        if self.kind == OPT_SGD:
            w.add_(g, alpha=-self.lr)
        elif self.kind == OPT_ADAM:
            t = self.step_count
            self.m.mul_(self.b1).add_(g, alpha=1 - self.b1)
            self.v.mul_(self.b2).addcmul_(g, g, value=1 - self.b2)
            bc1 = 1 - self.b1 ** t
            bc2 = 1 - self.b2 ** t
            # update = lr/bc1 * m / (sqrt(v)/sqrt(bc2) + eps)  (torch-Adam form)
            step = self.lr * math.sqrt(bc2) / bc1
            denom = self.v.sqrt().add_(self.eps * math.sqrt(bc2))
            w.addcdiv_(self.m, denom, value=-step)
        elif self.kind == OPT_ADADELTA:
            # TF AdadeltaOptimizer semantics (rho, eps inside both sqrts)
            self.m.mul_(self.rho).addcmul_(g, g, value=1 - self.rho)
            upd = g * (self.v.add(self.eps).sqrt_() / self.m.add(self.eps).sqrt_())
            self.v.mul_(self.rho).addcmul_(upd, upd, value=1 - self.rho)
            w.add_(upd, alpha=-self.lr)
        elif self.kind == OPT_ADAGRAD:
            self.m.addcmul_(g, g, value=1.0)
            w.addcdiv_(g, self.m.sqrt().add(self.eps), value=-self.lr)

    def _dense_step_hip(self) -> bool:
        """Returns True when the kernel also refreshed the bf16 mirror
        (the cast is fused into the update pass — one less arena-wide
        copy per step)."""
        ext = hip_ops()
        w, g = self.flat.flat, self.flat.flat_grad
        mir = self.flat.mirror
        if mir is None:
            mir = torch.empty(0, device=w.device, dtype=torch.bfloat16)
        if self.kind == OPT_SGD:
            ext.sgd_step(w, g, mir, self.lr, self.l2)
        elif self.kind == OPT_ADAM:
            # device-side step counter -> hipGraph-replayable bias correction
            if not hasattr(self, "_step_buf"):
                self._step_buf = torch.zeros(1, device=w.device)
                self._step_buf.fill_(float(self.step_count - 1))
            ext.adam_step_dev(w, g, self.m, self.v, mir, self._step_buf,
                              self.lr, self.b1, self.b2, self.eps, self.l2)
        elif self.kind == OPT_ADADELTA:
            ext.adadelta_step(w, g, self.m, self.v, mir, self.lr, self.rho,
                              self.eps, self.l2)
        elif self.kind == OPT_ADAGRAD:
            ext.adagrad_step(w, g, self.m, mir, self.lr, self.eps, self.l2)
        return True

    # ----------------------------------------------------------------- sparse
    def _emb_step(self, p: torch.nn.Parameter, idx: int) -> None:
        adagrad = self.emb_kind == OPT_ADAGRAD
        in_arena = bool(getattr(p, "_acc_in_arena", False))
        stash = getattr(p, "_unified_grads", None)
        if stash:
            # deferred unified-arena grads (ops/embedding._UnifiedGatherFn
            # fast path, GPU only): consume the unpacked buffers directly
            ext = hip_ops()
            acc = (self.emb_state[idx] if (adagrad and not in_arena)
                   else torch.empty(0, device=p.device, dtype=torch.float32))
            for (rows, dout, nd, dwide, F, _D) in stash:
                ext.emb_update_unified(p.data, acc, rows, dout, nd,
                                       dwide, 1, F, self.emb_lr, self.eps,
                                       adagrad, in_arena)
            p._unified_grads = []
            return
        if p.grad is None:
            return
        from shifu_amd.ops.embedding import sparse_rows_values
        rows, vals = sparse_rows_values(p.grad)
        if rows.numel() == 0:
            return
        if in_arena:
            self._emb_step_in_arena(p, rows, vals, adagrad)
        elif adagrad:
            acc = self.emb_state[idx]
            if p.dtype == torch.bfloat16 and use_hip(p):
                hip_ops().emb_adagrad_step(p.data, acc, rows.contiguous(),
                                           vals.contiguous(), self.emb_lr, self.eps)
            else:
                vf = vals.float()
                rowsq = (vf * vf).mean(dim=1)
                acc.index_add_(0, rows, rowsq)
                denom = acc[rows].add(self.eps).sqrt_().unsqueeze(1)
                p.data.index_add_(0, rows, (-self.emb_lr * vf / denom).to(p.dtype))
        else:  # sgd
            if p.dtype == torch.bfloat16 and use_hip(p):
                hip_ops().emb_sgd_step(p.data, rows.contiguous(), vals.contiguous(),
                                       self.emb_lr)
            else:
                p.data.index_add_(0, rows, (-self.emb_lr * vals.float()).to(p.dtype))
        p.grad = None

    def _emb_step_in_arena(self, p, rows, vals, adagrad: bool) -> None:
        """Packed [n, D+4] grads on a D+4 unified arena.  The generic kernels
        must NOT touch the accumulator columns (a packed-bf16 add of 0 can
        canonicalize a NaN bit pattern inside the raw f32), so the packed
        values route through the unified kernels: dgrad=vals, dcol0=0,
        wide=vals[:, D] (stride DP)."""
        DP = p.shape[1]
        D = DP - 4
        vals = vals.contiguous()
        if p.dtype == torch.bfloat16 and use_hip(p):
            hip_ops().emb_update_unified(
                p.data, torch.empty(0, device=p.device, dtype=torch.float32),
                rows.contiguous(), vals, 0, vals[:, D], DP, 1,
                self.emb_lr, self.eps, adagrad, True)
        else:
            vf = vals.float()
            gsq = (vf[:, :D] * vf[:, :D]).sum(dim=1) + vf[:, D] * vf[:, D]
            upd = torch.zeros_like(vf)
            if adagrad:
                acc = p.data[:, D + 2]          # fp32 arena: plain f32 col
                acc.index_add_(0, rows, gsq / DP)
                denom = acc[rows].add(self.eps).sqrt_().unsqueeze(1)
                upd[:, :D + 1] = -self.emb_lr * vf[:, :D + 1] / denom
            else:
                upd[:, :D + 1] = -self.emb_lr * vf[:, :D + 1]
            p.data.index_add_(0, rows, upd.to(p.dtype))
        p.grad = None

    # ------------------------------------------------------------------ api
    @torch.no_grad()
    def step(self) -> None:
        self.step_count += 1
        if self.flat.numel():
            if self.flat.flat.is_cuda:
                from shifu_amd.ops.linear import drain_wgrad_events
                drain_wgrad_events()
            self.flat.sync_grads()
            if use_hip(self.flat.flat):
                self._dense_step_hip()   # bf16 mirror refreshed in-kernel
            else:
                self._dense_step_ref()
                self.flat.refresh_mirror()
        for i, p in enumerate(self.emb_params):
            self._emb_step(p, i)

    def zero_grad(self) -> None:
        self.flat.zero_grad()
        for p in self.emb_params:
            p.grad = None

    # ------------------------------------------------------------ checkpoint
    def state_dict(self) -> dict:
        return {
            "step_count": self.step_count,
            "m": self.m, "v": self.v,
            "emb_state": self.emb_state,
        }

    def load_state_dict(self, sd: dict) -> None:
        self.step_count = int(sd["step_count"])
        if hasattr(self, "_step_buf"):
            self._step_buf.fill_(float(self.step_count))
        if self.m is not None and sd.get("m") is not None:
            self.m.copy_(sd["m"])
        if self.v is not None and sd.get("v") is not None:
            self.v.copy_(sd["v"])
        for k, t in (sd.get("emb_state") or {}).items():
            if int(k) in self.emb_state:
                self.emb_state[int(k)].copy_(t)
