"""Long-context CP attention bench on one MI355X (world 1): merged-flash
CP attention (prefix+diagonal LSE merge) vs plain flash at 32k-128k.

The world-1 path is plain flash; the world>1 merge path is exercised by
pinning prefix_len explicitly. Measures fwd+bwd ms and peak memory —
the point of the flash merge is that 128k contexts run at flash memory
(the round-1 fp32-scores version OOMed by construction).
"""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from torchft_amd.parallel.cp import _MergedFlashAttn

def bench(S_local, prefix, iters=5):
    B, Hq, Hkv, D = 1, 32, 8, 128
    dev = "cuda"
    q = torch.randn(B, Hq, S_local, D, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn(B, Hkv, prefix + S_local, D, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)
    v = torch.randn_like(k, requires_grad=True)
    g = torch.randn(B, Hq, S_local, D, device=dev, dtype=torch.bfloat16)
    def step():
        out = _MergedFlashAttn.apply(q, k, v, prefix)
        out.backward(g)
        q.grad = k.grad = v.grad = None
    for _ in range(2): step()
    torch.cuda.synchronize(); torch.cuda.reset_peak_memory_stats()
    t0 = time.perf_counter()
    for _ in range(iters): step()
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / iters * 1000
    peak = torch.cuda.max_memory_allocated() / 2**30
    print(f"S_local={S_local} prefix={prefix} (ctx={prefix+S_local}): "
          f"{ms:8.2f} ms fwd+bwd, peak {peak:.1f} GiB", flush=True)

if __name__ == "__main__":
    for S_local, prefix in [(16384, 16384), (32768, 32768), (65536, 65536)]:
        bench(S_local, prefix)
