"""Drive only the custom FA bwd kernels a few times (PMC profiling target)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torchft_amd.ops import hip_ext

B, Hq, Hkv, S, D = 2, 32, 8, 8192, 128
dev = "cuda"
torch.manual_seed(0)
q = torch.randn(B, Hq, S, D, device=dev, dtype=torch.bfloat16)
k = torch.randn(B, Hkv, S, D, device=dev, dtype=torch.bfloat16)
v = torch.randn(B, Hkv, S, D, device=dev, dtype=torch.bfloat16)
dout = torch.randn_like(q)
out, lse, *_ = torch.ops.aten._scaled_dot_product_flash_attention(
    q, k, v, 0.0, True, False, scale=D ** -0.5)
delta = hip_ext().fa_delta(dout, out)
for _ in range(3):
    hip_ext().fa_bwd(q, k, v, dout, lse, delta, D ** -0.5, True)
    hip_ext().fa_fwd(q, k, v, D ** -0.5, True)
torch.cuda.synchronize()
print("done")
