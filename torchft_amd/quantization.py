"""fp8 (OCP e4m3) block quantization for gradient collectives.

Thin tensor-level wrapper over the CDNA4 kernels in
``csrc/kernels/fp8_quant.hip`` plus a pure-torch reference implementation of
the exact packed layout (ground truth for the GPU numerics tests and a CPU
way to reason about the format).

Packed wire layout (per all-reduce world of size W): W equal slices; slice r
= ``[fp32 dequant-scales of its blocks] ‖ [QBLOCK fp8 bytes per block]``.
Blocks are 2048 elements; each tensor is padded to whole blocks so a block
never straddles tensors. Reference analogue: torchft/quantization.py
(Triton, per-row striping) — re-designed block-wise for MI355X (see the .hip
header comment).
"""

from __future__ import annotations

from typing import List, Tuple

import torch

from torchft_amd.ops import hip_ext

QBLOCK = 2048
FP8_MAX = 448.0


def pack_geometry(tensors: List[torch.Tensor], world: int) -> Tuple[int, int, int, int]:
    """Returns (total_blocks, padded_blocks, blocks_per_rank, slice_bytes)."""
    total = sum((t.numel() + QBLOCK - 1) // QBLOCK for t in tensors)
    padded = ((total + world - 1) // world) * world
    bpr = padded // world
    slice_bytes = bpr * (4 + QBLOCK)
    return total, padded, bpr, slice_bytes


def allocate_pack(tensors: List[torch.Tensor], world: int) -> torch.Tensor:
    _, _, _, slice_bytes = pack_geometry(tensors, world)
    return torch.empty(world * slice_bytes, dtype=torch.uint8, device=tensors[0].device)


def quantize_pack(tensors: List[torch.Tensor], pack: torch.Tensor, world: int) -> None:
    hip_ext().fp8_quantize(tensors, pack, world)


def dequantize_pack(tensors: List[torch.Tensor], pack: torch.Tensor, world: int) -> None:
    hip_ext().fp8_dequantize(tensors, pack, world)


def reduce_slices(recv: torch.Tensor, out: torch.Tensor, world: int, avg: bool) -> None:
    hip_ext().fp8_reduce(recv, out, world, avg)


# --------------------------------------------------------------- reference


def _f8_roundtrip(x: torch.Tensor) -> torch.Tensor:
    """Quantize fp32 → OCP e4m3 → fp32 using torch's float8 dtype."""
    return x.to(torch.float8_e4m3fn).float()


def quantize_pack_ref(tensors: List[torch.Tensor], world: int) -> torch.Tensor:
    """Pure-torch construction of the packed wire buffer (CPU or GPU)."""
    total, padded, bpr, slice_bytes = pack_geometry(tensors, world)
    pack = torch.zeros(world * slice_bytes, dtype=torch.uint8, device=tensors[0].device)

    blocks = []
    for t in tensors:
        flat = t.detach().float().reshape(-1)
        nb = (flat.numel() + QBLOCK - 1) // QBLOCK
        padded_flat = torch.zeros(nb * QBLOCK, dtype=torch.float32, device=flat.device)
        padded_flat[: flat.numel()] = flat
        blocks.append(padded_flat.view(nb, QBLOCK))
    allb = torch.cat(blocks)  # [total, QBLOCK]

    for b in range(padded):
        r, bl = divmod(b, bpr)
        sl = pack[r * slice_bytes : (r + 1) * slice_bytes]
        if b < total:
            blk = allb[b]
            amax = blk.abs().max()
            dq = (amax / FP8_MAX) if amax > 0 else torch.zeros(())
            q = (
                _f8_roundtrip(blk * (FP8_MAX / amax)).to(torch.float8_e4m3fn)
                if amax > 0
                else torch.zeros(QBLOCK, dtype=torch.float8_e4m3fn, device=blk.device)
            )
        else:
            dq = torch.zeros(())
            q = torch.zeros(QBLOCK, dtype=torch.float8_e4m3fn, device=allb.device)
        import struct

        sl[bl * 4 : (bl + 1) * 4] = torch.tensor(
            list(struct.pack("<f", float(dq))), dtype=torch.uint8, device=sl.device
        )
        payload_off = bpr * 4 + bl * QBLOCK
        sl[payload_off : payload_off + QBLOCK] = q.view(torch.uint8)
    return pack


def dequantize_pack_ref(
    shapes_numels: List[int], pack: torch.Tensor, world: int
) -> List[torch.Tensor]:
    """Decode a packed buffer into flat fp32 tensors of the given numels."""
    import struct

    total = sum((n + QBLOCK - 1) // QBLOCK for n in shapes_numels)
    padded = ((total + world - 1) // world) * world
    bpr = padded // world
    slice_bytes = bpr * (4 + QBLOCK)

    vals = []
    for b in range(total):
        r, bl = divmod(b, bpr)
        sl = pack[r * slice_bytes : (r + 1) * slice_bytes]
        dq = struct.unpack("<f", bytes(sl[bl * 4 : (bl + 1) * 4].cpu().tolist()))[0]
        payload_off = bpr * 4 + bl * QBLOCK
        q = sl[payload_off : payload_off + QBLOCK].view(torch.float8_e4m3fn).float()
        vals.append(q * dq)
    allb = torch.cat(vals) if vals else torch.zeros(0)

    out = []
    off = 0
    for n in shapes_numels:
        nb = (n + QBLOCK - 1) // QBLOCK
        out.append(allb[off : off + n].clone())
        off += nb * QBLOCK
    return out
