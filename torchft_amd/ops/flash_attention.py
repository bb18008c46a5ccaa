"""Flash attention with a hand-written CDNA4 backward.

Forward: aten's flash kernel (fast on gfx950, and it hands us the
logsumexp). Backward: the MFMA kernels in csrc/kernels/flash_attn_bwd.hip —
the stock backward is the single largest kernel of the Llama-8B step
(profiles/SUMMARY.md).

Constraints of the custom backward: bf16, head_dim=128, seq % 128 == 0,
dropout 0. Anything else falls back to stock SDPA.

Status (end of round 2, measured on MI355X at the 8B bench shape,
interleaved A/B): fwd+bwd 11.80 ms custom vs 13.27 ms stock (-11%) — the
hand-written backward beats aotriton's on both kernels, so it is ON by
default (opt out with TORCHFT_AMD_CUSTOM_FA=0). Levers in order: XOR
swizzle of the LDS images (21.2 -> 17.6 ms), ds_read_b64_tr_b16
hardware-transpose reads replacing the b16 transpose-scatter images
(-> 16.0), 8-wave workgroups for 2 waves/SIMD occupancy (-> 12.7), native
[B,S,H,D] layout + fused delta (-> 12.5 and no aten copy chain), 64-key
dq tiles (-> 11.8). The hand-written FORWARD (fa_fwd) is numerically
pinned against aten but ~0.65 ms behind it, so it stays opt-in
(TORCHFT_AMD_CUSTOM_FA_FWD=1; full-custom 12.4 ms still beats stock).
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.nn.functional as F

from torchft_amd.ops import hip_ext

_ENV = "TORCHFT_AMD_CUSTOM_FA"
_ENV_FWD = "TORCHFT_AMD_CUSTOM_FA_FWD"


def custom_fa_enabled() -> bool:
    v = os.environ.get(_ENV, "1")
    return v in ("1", "true", "True")


def custom_fa_fwd_enabled() -> bool:
    """The hand-written forward (opt-in until it beats aotriton's)."""
    v = os.environ.get(_ENV_FWD, "0")
    return v in ("1", "true", "True")


class _FlashAttentionFn(torch.autograd.Function):
    """q,k,v: [B, H, S, D] (SDPA layout), causal, GQA-native."""

    @staticmethod
    def forward(ctx, q, k, v, causal: bool, scale: float):
        if custom_fa_fwd_enabled() and q.shape[2] % 256 == 0:
            out, lse = hip_ext().fa_fwd(q, k, v, scale, causal)
        else:
            out, lse, *_ = torch.ops.aten._scaled_dot_product_flash_attention(
                q, k, v, 0.0, causal, False, scale=scale
            )
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.causal = causal
        ctx.scale = scale
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse = ctx.saved_tensors
        # D_i = rowsum(dout * out), fused fp32 kernel (no fp32 temporaries;
        # handles the model's [B,S,H,D]-backed transpose views natively)
        delta = hip_ext().fa_delta(dout, out)
        dq, dk, dv = hip_ext().fa_bwd(
            q, k, v, dout, lse, delta, ctx.scale, ctx.causal
        )
        return dq, dk, dv, None, None


def _supported(q: torch.Tensor, k: torch.Tensor) -> bool:
    return (
        q.is_cuda
        and q.dtype == torch.bfloat16
        and q.shape[-1] == 128
        and q.shape[2] % 128 == 0
        and k.shape[2] == q.shape[2]
        and q.shape[1] % k.shape[1] == 0
        and custom_fa_enabled()
        and hip_ext() is not None
    )


def flash_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    causal: bool = True,
    scale: Optional[float] = None,
) -> torch.Tensor:
    """SDPA-compatible attention ([B, H, S, D]) with the custom backward
    where supported, stock SDPA otherwise."""
    if scale is None:
        scale = q.shape[-1] ** -0.5
    if _supported(q, k):
        # no .contiguous(): the kernels consume both the packed [B,H,S,D]
        # layout and the model's [B,S,H,D]-backed transpose views natively
        return _FlashAttentionFn.apply(q, k, v, causal, scale)
    return F.scaled_dot_product_attention(
        q, k, v, is_causal=causal, scale=scale, enable_gqa=q.shape[1] != k.shape[1]
    )
