"""Llama-3 model family for the flagship fault-tolerant training benchmark.

New scope relative to the reference (torchft delegates the model stack to
torchtitan; BASELINE.json names Llama-3 8B/70B as the benchmark configs).
MI355X-first: bf16 parameters, fused CDNA4 RMSNorm/RoPE/SwiGLU kernels
(torchft_amd.ops), SDPA flash attention with GQA, chunked cross-entropy so
the 128k-vocab logits never materialize at once, and optional per-block
activation checkpointing sized for 288 GB HBM3E.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from torchft_amd.ops import RMSNorm, rope, rope_tables, swiglu_glu
from torchft_amd.ops.flash_attention import flash_attention


@dataclass
class CPPlan:
    """Context-parallel plan: this rank holds sequence shard ``rank`` of
    ``world``; attention all-gathers KV over ``pg`` (parallel/cp.py)."""

    pg: object  # torchft_amd.process_group.ProcessGroup
    rank: int
    world: int


@dataclass
class LlamaConfig:
    dim: int = 4096
    n_layers: int = 32
    n_heads: int = 32
    n_kv_heads: int = 8
    ffn_hidden: int = 14336
    vocab_size: int = 128256
    max_seq_len: int = 8192
    rope_theta: float = 500000.0
    norm_eps: float = 1e-5

    @property
    def head_dim(self) -> int:
        return self.dim // self.n_heads


LLAMA3_8B = LlamaConfig()
LLAMA3_70B = LlamaConfig(
    dim=8192, n_layers=80, n_heads=64, n_kv_heads=8, ffn_hidden=28672
)
LLAMA_DEBUG = LlamaConfig(
    dim=256, n_layers=2, n_heads=4, n_kv_heads=2, ffn_hidden=512,
    vocab_size=512, max_seq_len=512,
)


class Attention(nn.Module):
    def __init__(
        self, cfg: LlamaConfig, dtype: torch.dtype, cp: Optional[CPPlan] = None
    ) -> None:
        super().__init__()
        self.cfg = cfg
        self.cp = cp
        d, hd = cfg.dim, cfg.head_dim
        # fused QKV projection: one [d, (Hq+2*Hkv)*hd] GEMM instead of
        # three — fewer launches and a wider N for the MFMA pipeline
        self.wqkv = nn.Linear(
            d, (cfg.n_heads + 2 * cfg.n_kv_heads) * hd, bias=False, dtype=dtype
        )
        self.wo = nn.Linear(cfg.n_heads * hd, d, bias=False, dtype=dtype)

    def forward(self, x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
        B, S, _ = x.shape
        cfg = self.cfg
        hd = cfg.head_dim
        qkv = self.wqkv(x)
        nq = cfg.n_heads * hd
        nkv = cfg.n_kv_heads * hd
        q = qkv[..., :nq].view(B, S, cfg.n_heads, hd)
        k = qkv[..., nq : nq + nkv].view(B, S, cfg.n_kv_heads, hd)
        v = qkv[..., nq + nkv :].view(B, S, cfg.n_kv_heads, hd)
        # cos/sin already sliced to this rank's global positions under CP
        q = rope(q, cos, sin)
        k = rope(k, cos, sin)
        if self.cp is not None and self.cp.world > 1:
            from torchft_amd.parallel.cp import cp_attention

            out = cp_attention(
                q, k, v, self.cp.pg, self.cp.rank, self.cp.world, causal=True
            )
            return self.wo(out.reshape(B, S, -1))
        # SDPA layout [B, H, S, D]; custom CDNA4 backward where enabled
        # (stock SDPA handles the transposed views without a copy)
        q, k, v = (t.transpose(1, 2) for t in (q, k, v))
        out = flash_attention(q, k, v, causal=True)
        out = out.transpose(1, 2).reshape(B, S, -1)
        return self.wo(out)


class MLP(nn.Module):
    def __init__(self, cfg: LlamaConfig, dtype: torch.dtype) -> None:
        super().__init__()
        # fused gate+up: one [d, 2*ffn] GEMM feeding the fused SwiGLU kernel
        self.w13 = nn.Linear(cfg.dim, 2 * cfg.ffn_hidden, bias=False, dtype=dtype)
        self.w2 = nn.Linear(cfg.ffn_hidden, cfg.dim, bias=False, dtype=dtype)  # down

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.w2(swiglu_glu(self.w13(x)))


class Block(nn.Module):
    def __init__(
        self, cfg: LlamaConfig, dtype: torch.dtype, cp: Optional[CPPlan] = None
    ) -> None:
        super().__init__()
        self.attn_norm = RMSNorm(cfg.dim, cfg.norm_eps, dtype=dtype)
        self.attn = Attention(cfg, dtype, cp=cp)
        self.mlp_norm = RMSNorm(cfg.dim, cfg.norm_eps, dtype=dtype)
        self.mlp = MLP(cfg, dtype)

    def forward(self, x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
        x = x + self.attn(self.attn_norm(x), cos, sin)
        x = x + self.mlp(self.mlp_norm(x))
        return x


class Llama(nn.Module):
    def __init__(
        self,
        cfg: LlamaConfig,
        dtype: torch.dtype = torch.bfloat16,
        checkpoint_activations: bool = True,
        cp: Optional[CPPlan] = None,
    ) -> None:
        super().__init__()
        self.cfg = cfg
        self.cp = cp
        self.checkpoint_activations = checkpoint_activations
        self.tok_embeddings = nn.Embedding(cfg.vocab_size, cfg.dim, dtype=dtype)
        self.layers = nn.ModuleList(
            Block(cfg, dtype, cp=cp) for _ in range(cfg.n_layers)
        )
        self.norm = RMSNorm(cfg.dim, cfg.norm_eps, dtype=dtype)
        self.output = nn.Linear(cfg.dim, cfg.vocab_size, bias=False, dtype=dtype)

        cos, sin = rope_tables(cfg.max_seq_len, cfg.head_dim, cfg.rope_theta)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)

        self._init_weights()

    def _init_weights(self) -> None:
        std = 0.02
        for m in self.modules():
            if isinstance(m, nn.Linear):
                nn.init.normal_(m.weight, mean=0.0, std=std)
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, mean=0.0, std=std)

    def forward_hidden(self, tokens: torch.Tensor) -> torch.Tensor:
        """Token ids [B, S] → final hidden states [B, S, dim].

        Under CP, ``tokens`` is this rank's contiguous sequence shard and
        the rotary tables are sliced to its global positions."""
        x = self.tok_embeddings(tokens)
        if self.cp is not None and self.cp.world > 1:
            s = tokens.shape[1]
            off = self.cp.rank * s
            cos, sin = self.rope_cos[off : off + s], self.rope_sin[off : off + s]
        else:
            cos, sin = self.rope_cos, self.rope_sin
        for layer in self.layers:
            if self.checkpoint_activations and self.training:
                x = torch.utils.checkpoint.checkpoint(
                    layer, x, cos, sin, use_reentrant=False
                )
            else:
                x = layer(x, cos, sin)
        return self.norm(x)

    def forward(self, tokens: torch.Tensor) -> torch.Tensor:
        """Full logits — use forward_loss for training (chunked CE)."""
        return self.output(self.forward_hidden(tokens))

    def forward_loss(
        self,
        tokens: torch.Tensor,
        targets: torch.Tensor,
        chunk_rows: int = 4096,
    ) -> torch.Tensor:
        """Mean cross-entropy with chunked logits: the [B*S, vocab] logits
        matrix (2 GB+ at 128k vocab) is computed ``chunk_rows`` rows at a
        time so peak memory stays bounded."""
        h = self.forward_hidden(tokens)
        h = h.reshape(-1, self.cfg.dim)
        t = targets.reshape(-1)
        n = h.shape[0]
        losses = []
        for i in range(0, n, chunk_rows):
            logits = self.output(h[i : i + chunk_rows])
            # no .float(): torch's CE softmax accumulates in fp32 for bf16
            # inputs anyway, and the explicit cast materialized a 2 GB fp32
            # logits copy per chunk (plus its gradient) at the 128k vocab
            losses.append(
                F.cross_entropy(logits, t[i : i + chunk_rows], reduction="sum").float()
            )
        return torch.stack(losses).sum() / n

    def num_params(self) -> int:
        return sum(p.numel() for p in self.parameters())
