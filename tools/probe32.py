import torch, sys
sys.path.insert(0, "/root/repo")
from shifu_amd.ops.dispatch import hip_ops
ext = hip_ops()
g = torch.Generator().manual_seed(0)
a = torch.randn(32, 16, generator=g).to(torch.bfloat16).cuda()
b = (torch.arange(32*16).float().reshape(32,16) * 0.01 + torch.randn(32,16,generator=g)).to(torch.bfloat16).cuda()  # asymmetric B
d = ext.mfma_probe32(a, b)
ref = a.float() @ b.float().t()
err = (d - ref).abs().max().item()
print("probe32 maxerr:", err, "OK" if err < 1e-2 * ref.abs().max().item() else "FAIL")
