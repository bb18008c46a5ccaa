"""CDNA4 fused ops: RMSNorm, RoPE, SwiGLU, fused AdamW.

On a HIP device these route through the in-tree gfx950 extension
(``torchft_amd/_hip_kernels.so``, sources in ``csrc/kernels/``) and FAIL
LOUDLY if it is missing — a silent eager fallback on the GPU would
invalidate every benchmark. On CPU they fall back to plain fp32 torch
reference implementations (used by the numerics tests as ground truth).
"""

from __future__ import annotations

import os
from typing import Dict, List, Optional

import torch

_EXT = None
_EXT_ERR: Optional[str] = None


def _load_ext():
    global _EXT, _EXT_ERR
    if _EXT is not None:
        return _EXT
    import importlib.util

    so = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                      "_hip_kernels.so")
    try:
        spec = importlib.util.spec_from_file_location("torchft_amd._hip_kernels", so)
        assert spec is not None and spec.loader is not None
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        _EXT = mod
    except Exception as e:  # noqa: BLE001
        _EXT_ERR = f"{type(e).__name__}: {e}"
        _EXT = None
    return _EXT


def hip_ext():
    """The HIP kernel extension; raises on a GPU host if unavailable."""
    ext = _load_ext()
    if ext is None and torch.cuda.is_available():
        raise RuntimeError(
            f"torchft_amd HIP kernel extension not available on a GPU host "
            f"(build with `python -m torchft_amd._build`): {_EXT_ERR}"
        )
    return ext


def have_hip_ext() -> bool:
    return _load_ext() is not None


# ----------------------------------------------------------------- RMSNorm


def rmsnorm_ref(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
    """fp32 reference: y = x * rsqrt(mean(x^2)+eps) * w."""
    xf = x.float()
    inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * inv * w.float()).to(x.dtype)


class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
        y, invrms = hip_ext().rmsnorm_fwd(x, w, eps)
        ctx.save_for_backward(x, w, invrms)
        return y

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        x, w, invrms = ctx.saved_tensors
        dx, dw = hip_ext().rmsnorm_bwd(dy, x, w, invrms)
        return dx, dw.to(w.dtype), None


def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    if x.is_cuda:
        return _RMSNormFn.apply(x.contiguous(), w, eps)
    # CPU fallback keeps autograd via plain torch ops
    inv = torch.rsqrt(x.float().pow(2).mean(-1, keepdim=True) + eps)
    return (x.float() * inv * w.float()).to(x.dtype)


class RMSNorm(torch.nn.Module):
    def __init__(self, hidden: int, eps: float = 1e-5, dtype=torch.bfloat16) -> None:
        super().__init__()
        self.weight = torch.nn.Parameter(torch.ones(hidden, dtype=dtype))
        self.eps = eps

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return rmsnorm(x, self.weight, self.eps)


# ----------------------------------------------------------------- RoPE


def rope_tables(
    seq_len: int, head_dim: int, theta: float = 500000.0, device="cpu"
) -> tuple[torch.Tensor, torch.Tensor]:
    """Llama-3 rotary tables: cos/sin [S, D/2] fp32 (rope theta 500k)."""
    inv_freq = 1.0 / (
        theta ** (torch.arange(0, head_dim, 2, device=device).float() / head_dim)
    )
    t = torch.arange(seq_len, device=device).float()
    freqs = torch.outer(t, inv_freq)  # [S, D/2]
    return freqs.cos(), freqs.sin()


def rope_ref(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
    """Reference rotate-half RoPE on [B, S, H, D]."""
    S, D = x.shape[1], x.shape[-1]
    c = cos[:S].view(1, S, 1, D // 2).float()
    s = sin[:S].view(1, S, 1, D // 2).float()
    x1, x2 = x.float().chunk(2, dim=-1)
    return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1).to(x.dtype)


class _RopeFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor):
        ctx.save_for_backward(cos, sin)
        return hip_ext().rope_apply(x, cos, sin, False)

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        cos, sin = ctx.saved_tensors
        return hip_ext().rope_apply(dy, cos, sin, True), None, None


def rope(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
    """Apply rotary embedding to [B, S, H, D] (fused on HIP)."""
    if x.is_cuda:
        return _RopeFn.apply(x, cos, sin)
    S, D = x.shape[1], x.shape[-1]
    c = cos[:S].view(1, S, 1, D // 2)
    s = sin[:S].view(1, S, 1, D // 2)
    x1, x2 = x.chunk(2, dim=-1)
    return torch.cat(
        [x1 * c.to(x.dtype) - x2 * s.to(x.dtype), x2 * c.to(x.dtype) + x1 * s.to(x.dtype)],
        dim=-1,
    )


# ----------------------------------------------------------------- SwiGLU


def swiglu_ref(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    return (torch.nn.functional.silu(a.float()) * b.float()).to(a.dtype)


class _SwiGLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a: torch.Tensor, b: torch.Tensor):
        ctx.save_for_backward(a, b)
        return hip_ext().swiglu_fwd(a, b)

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        a, b = ctx.saved_tensors
        da, db = hip_ext().swiglu_bwd(dy, a, b)
        return da, db


def swiglu(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    if a.is_cuda:
        return _SwiGLUFn.apply(a, b)
    return torch.nn.functional.silu(a) * b


class _SwiGLUGluFn(torch.autograd.Function):
    """Packed-GLU SwiGLU: gu = [..., 2f] with gate = gu[..., :f], up =
    gu[..., f:] (the fused w13 projection's output) — the kernel reads the
    halves strided so no contiguous() copies of the activations happen."""

    @staticmethod
    def forward(ctx, gu: torch.Tensor):
        ctx.save_for_backward(gu)
        return hip_ext().swiglu_glu_fwd(gu)

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        (gu,) = ctx.saved_tensors
        return hip_ext().swiglu_glu_bwd(dy, gu)


def swiglu_glu(gu: torch.Tensor) -> torch.Tensor:
    """silu(gu[..., :f]) * gu[..., f:] for a packed [..., 2f] projection."""
    if gu.is_cuda:
        return _SwiGLUGluFn.apply(gu)
    f = gu.shape[-1] // 2
    return torch.nn.functional.silu(gu[..., :f]) * gu[..., f:]


# ----------------------------------------------------------------- Fused AdamW


def _local_view(t: torch.Tensor) -> torch.Tensor:
    """The local shard of a DTensor (FSDP2 sharded params/grads), else t.
    The update runs on the shard in place — mathematically identical for
    elementwise optimizers, and what lets the HSDP path use the fused
    kernel."""
    try:
        from torch.distributed.tensor import DTensor
    except ImportError:  # pragma: no cover
        return t
    return t.to_local() if isinstance(t, DTensor) else t


class FusedAdamW(torch.optim.Optimizer):
    """Single-kernel multi-tensor AdamW for bf16 params (fp32 moments).

    One launch per step updates every parameter (csrc/kernels/fused_adamw.hip);
    falls back to torch's foreach AdamW path on CPU. DTensor params (the
    FSDP2 sharded path) are updated on their local shards.
    """

    def __init__(
        self,
        params,
        lr: float = 1e-3,
        betas=(0.9, 0.95),
        eps: float = 1e-8,
        weight_decay: float = 0.01,
    ) -> None:
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):  # noqa: D102
        assert closure is None
        for group in self.param_groups:
            # Params that intermittently have grad=None (e.g. unused-param
            # DDP) accumulate different step counts; the bias correction is
            # per step count, so bucket by it — one kernel launch per
            # distinct count (one in the common case).
            by_step: Dict[int, List[List[torch.Tensor]]] = {}
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                p_loc = _local_view(p)
                g_loc = _local_view(p.grad)
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p_loc, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(p_loc, dtype=torch.float32)
                state["step"] += 1
                bucket = by_step.setdefault(state["step"], [[], [], [], []])
                bucket[0].append(p_loc)
                bucket[1].append(g_loc)
                bucket[2].append(state["exp_avg"])
                bucket[3].append(state["exp_avg_sq"])

            beta1, beta2 = group["betas"]
            for step, (params, grads, exp_avgs, exp_avg_sqs) in by_step.items():
                if params[0].is_cuda and params[0].dtype == torch.bfloat16:
                    hip_ext().adamw_step(
                        params, grads, exp_avgs, exp_avg_sqs, group["lr"], beta1,
                        beta2, group["eps"], group["weight_decay"], step,
                    )
                else:
                    bc1 = 1 - beta1**step
                    bc2 = 1 - beta2**step
                    for p, g, m, v in zip(params, grads, exp_avgs, exp_avg_sqs):
                        gf = g.float()
                        m.mul_(beta1).add_(gf, alpha=1 - beta1)
                        v.mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
                        denom = (v / bc2).sqrt_().add_(group["eps"])
                        update = (m / bc1) / denom + group["weight_decay"] * p.float()
                        p.add_((-group["lr"] * update).to(p.dtype))
        return None
