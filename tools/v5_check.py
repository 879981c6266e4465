#!/usr/bin/env python3
"""v5 8-phase GEMM validation + A/B vs v4 (run on a GPU box).

Numerics: every EPI route (plain bf16, bias+act, f32 split-K) against fp32
torch references on transpose-detecting random inputs, including edge shapes
(partial tiles in M/N, K not a multiple of 64 -> guarded staging paths).
Perf: within-process A/B on the bench-relevant shapes.
"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
from shifu_amd.ops.dispatch import hip_ops


def relerr(a, b):
    return float((a.float() - b.float()).abs().max() /
                 b.float().abs().max().clamp_min(1e-6))


def check(ext):
    torch.manual_seed(0)
    bad = 0
    shapes = [(512, 512, 512), (4096, 4096, 4096), (512, 256, 264),
              (777, 300, 100), (512, 260, 130), (1000, 1024, 1864),
              (256, 256, 64), (300, 70, 40)]
    for (M, N, K) in shapes:
        a = (torch.randn(M, K, device="cuda") * 0.5).to(torch.bfloat16)
        b = (torch.randn(N, K, device="cuda") * 0.5).to(torch.bfloat16)
        bias = torch.randn(N, device="cuda").to(torch.bfloat16)
        ref = a.float() @ b.float().t()

        c1 = ext.gemm_ntv3_bf16(a, b)
        e1 = relerr(c1, ref)
        c2 = ext.gemm_ntv3_f32(a, b)
        e2 = relerr(c2, ref)
        c3 = ext.linear_nt_fwd(a, b, bias, 3)  # relu
        ref3 = torch.relu(ref + bias.float())
        e3 = relerr(c3, ref3)
        tol = 3e-2
        ok = e1 < tol and e2 < tol and e3 < tol
        bad += 0 if ok else 1
        print(f"  [{M}x{N}x{K}] nt={e1:.2e} f32={e2:.2e} fwd={e3:.2e} "
              f"{'OK' if ok else 'FAIL'}", flush=True)
    return bad


def bench(ext, name, M, N, K, kind, iters=40):
    g = torch.Generator(device="cuda").manual_seed(1)
    a = (torch.randn(M, K, device="cuda", generator=g) * 0.5).to(torch.bfloat16)
    b = (torch.randn(N, K, device="cuda", generator=g) * 0.5).to(torch.bfloat16)
    bias = torch.randn(N, device="cuda", generator=g).to(torch.bfloat16)
    if kind == "nt":
        fn = lambda: ext.gemm_ntv3_bf16(a, b)
    elif kind == "f32":
        fn = lambda: ext.gemm_ntv3_f32(a, b)
    else:
        fn = lambda: ext.linear_nt_fwd(a, b, bias, 3)
    for _ in range(6):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    us = (time.time() - t0) / iters * 1e6
    tf = 2.0 * M * N * K / (us * 1e-6) / 1e12
    return us, tf


def main():
    ext = hip_ops()
    assert torch.cuda.is_available()
    print(f"== numerics (SHIFU_GEMM_V5={os.environ.get('SHIFU_GEMM_V5','1')})",
          flush=True)
    bad = check(ext)
    print(f"numerics: {'ALL OK' if bad == 0 else f'{bad} FAILURES'}", flush=True)

    shapes = [("sq4096", 4096, 4096, 4096, "nt"),
              ("fwd_l1", 32768, 1024, 1864, "fwd"),
              ("fwd_l2", 32768, 512, 1024, "fwd"),
              ("dgrad1", 32768, 1864, 1024, "nt"),
              ("wgrad1", 1024, 1864, 32768, "f32"),
              ("sq8192", 8192, 8192, 8192, "nt")]
    print(f"{'shape':<8} {'us':>9} {'TF':>8}")
    for nm, M, N, K, kind in shapes:
        us, tf = bench(ext, nm, M, N, K, kind)
        print(f"{nm:<8} {us:9.1f} {tf:8.1f}", flush=True)


if __name__ == "__main__":
    main()
