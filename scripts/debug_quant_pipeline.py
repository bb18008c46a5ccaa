"""Single-threaded emulation of the world-2 quantized allreduce pipeline on
one GPU, compared step by step against the pure-torch q8 reference."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torchft_amd.ops import hip_ext
from torchft_amd import quantization as Q

def q8(x):
    nb = (x.numel() + Q.QBLOCK - 1) // Q.QBLOCK
    xp = torch.zeros(nb * Q.QBLOCK, device=x.device)
    xp[: x.numel()] = x.float()
    xb = xp.view(nb, Q.QBLOCK)
    amax = xb.abs().amax(1, keepdim=True).clamp_min(1e-30)
    q = (xb * (Q.FP8_MAX / amax)).to(torch.float8_e4m3fn).float() * (amax / Q.FP8_MAX)
    return q.view(-1)[: x.numel()]

dev = "cuda"
torch.manual_seed(11)
world = 2
n = 100_000
inputs = [torch.randn(n, device=dev, dtype=torch.bfloat16) for _ in range(world)]
_, _, bpr, slice_bytes = Q.pack_geometry([inputs[0]], world)
print(f"bpr={bpr} slice_bytes={slice_bytes}")

# per-rank quantize
packs = []
for r in range(world):
    p = torch.empty(world * slice_bytes, dtype=torch.uint8, device=dev)
    Q.quantize_pack([inputs[r].clone()], p, world)
    packs.append(p)
torch.cuda.synchronize()

# roundtrip check per rank
for r in range(world):
    dec = [torch.zeros(n, device=dev, dtype=torch.bfloat16)]
    Q.dequantize_pack(dec, packs[r], world)
    torch.cuda.synchronize()
    ref = q8(inputs[r]).to(torch.bfloat16)
    d = (dec[0].float() - ref.float()).abs().max().item()
    print(f"rank {r} quant roundtrip vs q8 ref: max diff {d:.6f}")

# emulate alltoall for each rank + reduce + allgather + dequant
expected = q8(sum(q8(t) for t in inputs))
for r in range(world):
    recv = torch.cat([p[r * slice_bytes:(r + 1) * slice_bytes] for p in packs])
    my = torch.empty(slice_bytes, dtype=torch.uint8, device=dev)
    Q.reduce_slices(recv, my, world, False)
    torch.cuda.synchronize()
    # allgather emulation needs every rank's reduced slice
    if r == 0:
        slices = []
    slices.append(my)
final = torch.cat(slices)
out = [torch.zeros(n, device=dev, dtype=torch.bfloat16)]
Q.dequantize_pack(out, final, world)
torch.cuda.synchronize()
d = (out[0].float() - expected.float()).abs()
print(f"pipeline vs double-q8 ref: max {d.max().item():.4f} mean {d.mean().item():.6f} "
      f">0.06: {(d > 0.06 + 0.06 * expected.abs()).sum().item()}")
exact = sum(t.float() for t in inputs)
d2 = (out[0].float() - exact).abs()
print(f"pipeline vs exact sum: max {d2.max().item():.4f}")
