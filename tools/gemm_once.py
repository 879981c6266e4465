#!/usr/bin/env python3
"""Run one GEMM shape repeatedly (clean PMC capture target)."""
import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
from shifu_amd.ops.dispatch import hip_ops
M, N, K, iters = (int(x) for x in (sys.argv[1:] + ["4096", "4096", "4096", "20"])[:4])
a = torch.randn(M, K, device="cuda").to(torch.bfloat16)
b = torch.randn(N, K, device="cuda").to(torch.bfloat16)
for _ in range(iters):
    c = hip_ops().gemm_ntv3_bf16(a, b)
torch.cuda.synchronize()
print("done", c.shape)
