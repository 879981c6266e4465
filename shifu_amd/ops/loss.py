"""Fused sigmoid-head + weighted loss (+ gradient).

The reference puts a 1-unit sigmoid head named `shifu_output_0` on the MLP
(ssgd_monitor.py:121) and trains with per-sample-weighted MSE
(tf.losses.mean_squared_error(pred, y, weights) — ssgd_monitor.py:129).
Here the head stays in LOGITS through the network and sigmoid is fused into
the loss kernel, which is both faster (one pass over
[B]-sized data) and numerically stable for the cross-entropy variant.

Kinds:
* "weighted_mse": L = sum_i w_i (sigmoid(z_i) - y_i)^2 / #{i: w_i != 0}
* "sigmoid_ce" : L = sum_i w_i BCE(sigmoid(z_i), y_i) / #{i: w_i != 0}

Both normalize by the COUNT of nonzero-weight samples — TF's
SUM_BY_NONZERO_WEIGHTS, the reference's tf.losses.mean_squared_error
default reduction — so training with a real weight column keeps the same
loss scale and effective learning rate as the reference.  With the all-ones
default weights this equals the plain mean.
"""
from __future__ import annotations

import torch

from shifu_amd.ops.dispatch import use_hip, hip_ops

LOSS_WMSE, LOSS_SIGMOID_CE = 0, 1
_KIND_IDS = {"weighted_mse": LOSS_WMSE, "wmse": LOSS_WMSE,
             "sigmoid_ce": LOSS_SIGMOID_CE, "ce": LOSS_SIGMOID_CE,
             "bce": LOSS_SIGMOID_CE}


class _WeightedLossFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, z: torch.Tensor, y: torch.Tensor, w: torch.Tensor, kind: int):
        z1 = z.reshape(-1)
        if use_hip(z1):
            ext = hip_ops()
            # third output = count of nonzero weights (SUM_BY_NONZERO_WEIGHTS)
            p, loss_sum, wsum = ext.weighted_loss_fwd(
                z1.contiguous(), y.contiguous(), w.contiguous(), kind)
            loss = loss_sum / wsum.clamp_min(1e-12)
        else:
            zf = z1.float()
            p = torch.sigmoid(zf)
            yf, wf = y.float(), w.float()
            if kind == LOSS_WMSE:
                per = wf * (p - yf) ** 2
            else:
                per = wf * torch.nn.functional.binary_cross_entropy_with_logits(
                    zf, yf, reduction="none")
            wsum = (wf != 0).float().sum()
            loss = per.sum() / wsum.clamp_min(1e-12)
        ctx.save_for_backward(p, y, w, wsum if torch.is_tensor(wsum) else torch.tensor(wsum))
        ctx.kind = kind
        ctx.z_shape = z.shape
        ctx.z_dtype = z.dtype
        ctx.hip = use_hip(z1)
        return loss

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        p, y, w, wsum = ctx.saved_tensors
        scale = grad_out / wsum.clamp_min(1e-12)
        if ctx.hip:
            ext = hip_ops()
            dz = ext.weighted_loss_bwd(p, y.contiguous(), w.contiguous(),
                                       ctx.kind,
                                       scale.reshape(1).float().contiguous())
        else:
            yf, wf = y.float(), w.float()
            if ctx.kind == LOSS_WMSE:
                dz = wf * 2.0 * (p - yf) * p * (1.0 - p) * scale
            else:
                dz = wf * (p - yf) * scale
        dz = dz.reshape(ctx.z_shape).to(ctx.z_dtype)
        return dz, None, None, None


def weighted_loss(logits: torch.Tensor, target: torch.Tensor,
                  weight: torch.Tensor, kind: str = "weighted_mse") -> torch.Tensor:
    """Scalar weighted loss over a batch of head logits [B] or [B,1]."""
    return _WeightedLossFn.apply(logits, target, weight, _KIND_IDS[kind.lower()])


@torch.no_grad()
def predict_proba(logits: torch.Tensor) -> torch.Tensor:
    return torch.sigmoid(logits.float().reshape(-1))
