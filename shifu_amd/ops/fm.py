"""Fused factorization-machine second-order interaction.

fm2[b] = 0.5 * sum_d [(sum_f v[b,f,d])^2 - sum_f v[b,f,d]^2]

One fused kernel each way on GPU (the eager form is ~10 elementwise/reduce
kernels in fp32); pure-torch reference on CPU.
"""
from __future__ import annotations

import torch

from shifu_amd.ops.dispatch import use_hip, hip_ops


class _FM2Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, emb_flat: torch.Tensor, F: int, D: int):
        B = emb_flat.shape[0]
        if (emb_flat.dtype == torch.bfloat16 and use_hip(emb_flat)
                and emb_flat.stride(1) == 1):
            fm2, s = hip_ops().fm2_fwd(emb_flat, F, D)   # strided view OK
            ctx.save_for_backward(emb_flat, s)
        else:
            v = emb_flat.reshape(B, F, D).float()
            s = v.sum(dim=1)
            fm2 = 0.5 * (s * s - (v * v).sum(dim=1)).sum(dim=1)
            ctx.save_for_backward(emb_flat, s)
        ctx.F, ctx.D = F, D
        ctx.hip = (emb_flat.dtype == torch.bfloat16 and use_hip(emb_flat)
                   and emb_flat.stride(1) == 1)
        return fm2

    @staticmethod
    def backward(ctx, dout: torch.Tensor):
        emb_flat, s = ctx.saved_tensors
        F, D = ctx.F, ctx.D
        B = emb_flat.shape[0]
        if ctx.hip:
            demb = hip_ops().fm2_bwd(emb_flat, s, dout.float().contiguous(), F, D)
        else:
            v = emb_flat.reshape(B, F, D).float()
            demb = ((s.unsqueeze(1) - v) * dout.reshape(B, 1, 1).float()) \
                .reshape(B, F * D).to(emb_flat.dtype)
        return demb, None, None


def fm_second_order(emb_flat: torch.Tensor, F: int, D: int) -> torch.Tensor:
    """[B, F*D] concatenated per-feature embeddings -> [B] interaction term."""
    return _FM2Fn.apply(emb_flat, F, D)
