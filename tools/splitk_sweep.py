#!/usr/bin/env python3
"""Time gemm_ntv3_f32 (the wgrad route) on the bench wgrad shapes under the
current SHIFU_SPLITK_TARGET / SHIFU_SPLITK_MINKT / SHIFU_DISABLE_V4 env.

The knobs are read once per process (static init), so sweep by re-running:
  for t in 256 512 1024 2048; do SHIFU_SPLITK_TARGET=$t python tools/splitk_sweep.py; done
Prints one line per shape: env, shape, us, TFLOP/s, maxrel vs fp32 torch.
"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch
from shifu_amd.ops.dispatch import hip_ops

# (name, M, N, K): C[M,N] f32 = A[M,K] @ B[N,K]^T — A=dz^T, B=x^T at batch 32768
SHAPES = [
    ("l1wg", 1024, 1864, 32768),
    ("l2wg", 512, 1024, 32768),
    ("l3wg", 256, 512, 32768),
]


def main():
    ext = hip_ops()
    assert ext is not None and torch.cuda.is_available()
    torch.manual_seed(0)
    env = {k: os.environ.get(k, "-") for k in
           ("SHIFU_SPLITK_TARGET", "SHIFU_SPLITK_MINKT", "SHIFU_DISABLE_V4")}
    tag = f"tgt={env['SHIFU_SPLITK_TARGET']} minkt={env['SHIFU_SPLITK_MINKT']} nov4={env['SHIFU_DISABLE_V4']}"
    iters = int(os.environ.get("SWEEP_ITERS", "50"))
    for name, M, N, K in SHAPES:
        a = torch.randn(M, K, device="cuda").to(torch.bfloat16)
        b = torch.randn(N, K, device="cuda").to(torch.bfloat16)
        want = a.float() @ b.float().t()
        got = ext.gemm_ntv3_f32(a, b)
        rel = float((got - want).abs().max() / want.abs().max())
        for _ in range(5):
            ext.gemm_ntv3_f32(a, b)
        torch.cuda.synchronize()
        t0 = time.time()
        for _ in range(iters):
            ext.gemm_ntv3_f32(a, b)
        torch.cuda.synchronize()
        us = (time.time() - t0) / iters * 1e6
        tf = 2.0 * M * N * K / (us * 1e-6) / 1e12
        print(f"{tag}  {name} [{M},{N},{K}]  {us:8.1f} us  {tf:7.1f} TF  maxrel={rel:.2e}",
              flush=True)


if __name__ == "__main__":
    main()
