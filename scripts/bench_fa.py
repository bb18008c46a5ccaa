"""A/B: custom flash-attention backward vs stock SDPA backward at the
Llama-8B bench shape. Run on a GPU box."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F
from torchft_amd.ops.flash_attention import _FlashAttentionFn

B, Hq, Hkv, S, D = 2, 32, 8, 8192, 128
dev = "cuda"
torch.manual_seed(0)
q0 = torch.randn(B, Hq, S, D, device=dev, dtype=torch.bfloat16)
k0 = torch.randn(B, Hkv, S, D, device=dev, dtype=torch.bfloat16)
v0 = torch.randn(B, Hkv, S, D, device=dev, dtype=torch.bfloat16)
dout = torch.randn(B, Hq, S, D, device=dev, dtype=torch.bfloat16)

def run(custom: bool, iters=10, custom_fwd=False):
    q = q0.clone().requires_grad_(True)
    k = k0.clone().requires_grad_(True)
    v = v0.clone().requires_grad_(True)
    os.environ["TORCHFT_AMD_CUSTOM_FA_FWD"] = "1" if custom_fwd else "0"
    def step():
        if custom:
            out = _FlashAttentionFn.apply(q, k, v, True, D ** -0.5)
        else:
            out = F.scaled_dot_product_attention(q, k, v, is_causal=True, enable_gqa=True)
        out.backward(dout)
        q.grad = k.grad = v.grad = None
    for _ in range(3): step()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): step()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000

# interleaved A/B (guide rule 24)
import os
ROUNDS = int(os.environ.get("FA_ROUNDS", "3"))
for rnd in range(ROUNDS):
    a = run(False); b = run(True); c = run(True, custom_fwd=True)
    fl_fwd = 2*2*B*Hq*S*S*D/2
    fl_bwd = fl_fwd * 2.5
    print(f"round {rnd}: custom-fwd+bwd {c:.2f} ms", flush=True)
    print(f"round {rnd}: stock {a:.2f} ms | custom {b:.2f} ms "
          f"(fwd+bwd {B*Hq=}, eff stock {(fl_fwd+fl_bwd)/a/1e9:.0f} TF/s, "
          f"custom {(fl_fwd+fl_bwd)/b/1e9:.0f} TF/s)", flush=True)
