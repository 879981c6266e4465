#!/usr/bin/env python3
"""GEMM kernel microbenchmark on MI355X: our HIP kernels vs torch (rocBLAS)
on the bench-relevant shapes + a 4096^3 reference point.

Usage (on a GPU box): python tools/gemm_bench.py [--iters 50]
"""
import argparse
import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch
from shifu_amd.ops.dispatch import hip_ops

SHAPES = [
    # (name, M, N, K, kind)  kind: nn|nt|tn|v3|v3f  (C[M,N], reduction K)
    ("l1_fwd", 8192, 1024, 1864, "nn"),
    ("l1_fwd3", 8192, 1024, 1864, "v3"),
    ("l2_fwd3", 8192, 512, 1024, "v3"),
    ("l1_dgrad", 8192, 1864, 1024, "nt"),
    ("l1_dgrd3", 8192, 1864, 1024, "v3"),
    ("l1_wgrad", 1864, 1024, 8192, "tn"),
    ("l1_wgrd3", 1024, 1864, 8192, "v3f"),
    ("l1_wgrdT", 1024, 1864, 8192, "tt"),
    ("l1wgT16k", 1024, 1864, 16384, "tt"),
    ("l1_wgr3T", 1024, 1864, 8192, "tt3"),
    ("l1wg32k", 1024, 1864, 32768, "v3f"),
    ("l1wg32kT", 1024, 1864, 32768, "tt3"),
    ("l2wg32k", 512, 1024, 32768, "v3f"),
    ("l2wg32kT", 512, 1024, 32768, "tt3"),
    ("l3wg32kT", 256, 512, 32768, "tt3"),
    ("l3wg32k", 256, 512, 32768, "v3f"),
    ("l1f16k", 16384, 1024, 1864, "v3"),
    ("l1d16k", 16384, 1864, 1024, "v3"),
    ("sq4096", 4096, 4096, 4096, "nn"),
    ("sq4096v3", 4096, 4096, 4096, "v3"),
]


def run(fn, iters):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    import time
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=30)
    args = ap.parse_args()
    ext = hip_ops()
    assert ext is not None and torch.cuda.is_available()
    torch.manual_seed(0)
    print(f"{'shape':<10} {'kind':<4} {'ours_us':>8} {'ours_TF':>8} {'torch_us':>9} {'torch_TF':>9} {'maxrel':>8}")
    for name, M, N, K, kind in SHAPES:
        flops = 2.0 * M * N * K
        if kind == "nn":
            a = torch.randn(M, K, device="cuda").to(torch.bfloat16)
            b = torch.randn(K, N, device="cuda").to(torch.bfloat16)
            ours = lambda: ext.gemm_nn_bf16(a, b)
            ref = lambda: a @ b
        elif kind == "nt":
            a = torch.randn(M, K, device="cuda").to(torch.bfloat16)   # dz [B,N']
            b = torch.randn(N, K, device="cuda").to(torch.bfloat16)   # w [K',N']
            ours = lambda: ext.gemm_nt_bf16(a, b)
            ref = lambda: a @ b.t()
        elif kind == "tn":
            a = torch.randn(K, M, device="cuda").to(torch.bfloat16)   # x [B,K']
            b = torch.randn(K, N, device="cuda").to(torch.bfloat16)   # dz [B,N']
            ours = lambda: ext.gemm_tn_f32(a, b)
            ref = lambda: (a.t().float() @ b.float())
        elif kind == "v3":
            a = torch.randn(M, K, device="cuda").to(torch.bfloat16)
            b = torch.randn(N, K, device="cuda").to(torch.bfloat16)
            ours = lambda: ext.gemm_ntv3_bf16(a, b)
            ref = lambda: a @ b.t()
        elif kind == "tt":  # transpose-free wgrad: dz[B,M], x[B,N]
            a0 = torch.randn(K, M, device="cuda").to(torch.bfloat16)
            b0 = torch.randn(K, N, device="cuda").to(torch.bfloat16)
            ours = lambda: ext.gemm_tt_f32(a0, b0)
            ref = lambda: a0.t().float() @ b0.float()
            a, b = ext.transpose_bf16(a0), ext.transpose_bf16(b0)
        elif kind == "tt3":  # ttv3 transpose-free wgrad (scatter-staged v3)
            a0 = torch.randn(K, M, device="cuda").to(torch.bfloat16)
            b0 = torch.randn(K, N, device="cuda").to(torch.bfloat16)
            ours = lambda: ext.gemm_ttv3_f32(a0, b0)
            ref = lambda: a0.t().float() @ b0.float()
            a, b = ext.transpose_bf16(a0), ext.transpose_bf16(b0)
        else:  # v3f: wgrad incl. the two activation transposes
            a0 = torch.randn(K, M, device="cuda").to(torch.bfloat16)  # dz [B,N']
            b0 = torch.randn(K, N, device="cuda").to(torch.bfloat16)  # x [B,K']
            ours = lambda: ext.gemm_ntv3_f32(ext.transpose_bf16(a0), ext.transpose_bf16(b0))
            ref = lambda: a0.t().float() @ b0.float()
            a, b = ext.transpose_bf16(a0), ext.transpose_bf16(b0)

        c1 = ours().float()
        c2 = (a.float() @ b.float()) if kind == "nn" else \
             (a.float() @ b.float().t()) if kind in ("nt", "v3", "v3f", "tt", "tt3") else \
             (a.float().t() @ b.float())
        rel = float((c1 - c2).abs().max() / c2.abs().max().clamp_min(1e-3))

        t_ours = run(ours, args.iters)
        t_ref = run(ref, args.iters)
        print(f"{name:<10} {kind:<4} {t_ours*1e6:>8.1f} {flops/t_ours/1e12:>8.1f} "
              f"{t_ref*1e6:>9.1f} {flops/t_ref/1e12:>9.1f} {rel:>8.4f}")


if __name__ == "__main__":
    main()
