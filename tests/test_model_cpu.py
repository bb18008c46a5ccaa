"""CPU tests for the model stack (HIP-free fallback paths)."""

import torch

from torchft_amd.models.llama import LLAMA_DEBUG, Llama
from torchft_amd.ops import rmsnorm, rmsnorm_ref, rope, rope_ref, rope_tables, swiglu, swiglu_ref


class TestOpsFallbacks:
    def test_rmsnorm_cpu_matches_ref(self):
        torch.manual_seed(0)
        x = torch.randn(8, 64, dtype=torch.bfloat16)
        w = torch.randn(64, dtype=torch.bfloat16)
        torch.testing.assert_close(rmsnorm(x, w, 1e-5), rmsnorm_ref(x, w, 1e-5))

    def test_rope_cpu_matches_ref(self):
        torch.manual_seed(1)
        x = torch.randn(2, 16, 4, 32, dtype=torch.bfloat16)
        cos, sin = rope_tables(16, 32)
        torch.testing.assert_close(
            rope(x, cos, sin).float(), rope_ref(x, cos, sin).float(),
            rtol=2e-2, atol=2e-2,
        )

    def test_swiglu_cpu_matches_ref(self):
        torch.manual_seed(2)
        a = torch.randn(256, dtype=torch.bfloat16)
        b = torch.randn(256, dtype=torch.bfloat16)
        torch.testing.assert_close(
            swiglu(a, b).float(), swiglu_ref(a, b).float(), rtol=2e-2, atol=2e-2
        )


class TestLlamaCPU:
    def test_forward_backward_finite(self):
        torch.manual_seed(3)
        model = Llama(LLAMA_DEBUG, dtype=torch.float32, checkpoint_activations=False)
        toks = torch.randint(0, LLAMA_DEBUG.vocab_size, (2, 33))
        loss = model.forward_loss(toks[:, :-1], toks[:, 1:])
        loss.backward()
        assert torch.isfinite(loss)
        grads = [p.grad for p in model.parameters() if p.grad is not None]
        assert len(grads) > 0
        assert all(torch.isfinite(g).all() for g in grads)

    def test_chunked_ce_matches_full(self):
        torch.manual_seed(4)
        model = Llama(LLAMA_DEBUG, dtype=torch.float32, checkpoint_activations=False)
        toks = torch.randint(0, LLAMA_DEBUG.vocab_size, (1, 65))
        x, y = toks[:, :-1], toks[:, 1:]
        loss_chunked = model.forward_loss(x, y, chunk_rows=7)
        logits = model(x)
        loss_full = torch.nn.functional.cross_entropy(
            logits.reshape(-1, LLAMA_DEBUG.vocab_size).float(), y.reshape(-1)
        )
        torch.testing.assert_close(loss_chunked, loss_full, rtol=1e-5, atol=1e-5)

    def test_checkpointing_same_loss(self):
        torch.manual_seed(5)
        m1 = Llama(LLAMA_DEBUG, dtype=torch.float32, checkpoint_activations=False)
        torch.manual_seed(5)
        m2 = Llama(LLAMA_DEBUG, dtype=torch.float32, checkpoint_activations=True)
        m2.train()
        toks = torch.randint(0, LLAMA_DEBUG.vocab_size, (1, 33))
        l1 = m1.forward_loss(toks[:, :-1], toks[:, 1:])
        l2 = m2.forward_loss(toks[:, :-1], toks[:, 1:])
        torch.testing.assert_close(l1, l2)


class TestFusedAdamWCPUFallback:
    def test_matches_torch_adamw(self):
        from torchft_amd.ops import FusedAdamW

        torch.manual_seed(9)
        p = torch.randn(200, requires_grad=True)
        ref = p.detach().clone().requires_grad_(True)
        opt = FusedAdamW([p], lr=1e-2, betas=(0.9, 0.95), weight_decay=0.01)
        ref_opt = torch.optim.AdamW(
            [ref], lr=1e-2, betas=(0.9, 0.95), weight_decay=0.01, eps=1e-8
        )
        for _ in range(5):
            g = torch.randn_like(p)
            p.grad = g.clone()
            ref.grad = g.clone()
            opt.step()
            ref_opt.step()
        torch.testing.assert_close(p, ref, rtol=1e-5, atol=1e-6)


class TestFusedAdamWStepBuckets:
    def test_intermittent_grads_match_torch_adamw(self):
        """Params whose grads appear intermittently must get bias correction
        for their own step count, not the first param's."""
        from torchft_amd.ops import FusedAdamW

        torch.manual_seed(0)
        p_ref = [torch.randn(8, requires_grad=True) for _ in range(2)]
        p_fused = [p.detach().clone().requires_grad_(True) for p in p_ref]
        ref = torch.optim.AdamW(p_ref, lr=0.1, betas=(0.9, 0.95), eps=1e-8,
                                weight_decay=0.01)
        fused = FusedAdamW(p_fused, lr=0.1)

        for it in range(5):
            g0 = torch.randn(8)
            for opt_params, opt in ((p_ref, ref), (p_fused, fused)):
                opt_params[0].grad = g0.clone()
                # second param only gets a grad on even iterations
                opt_params[1].grad = (g0 * 2).clone() if it % 2 == 0 else None
                opt.step()
                opt.zero_grad(set_to_none=True)
        torch.testing.assert_close(p_fused[0], p_ref[0], rtol=1e-5, atol=1e-6)
        torch.testing.assert_close(p_fused[1], p_ref[1], rtol=1e-5, atol=1e-6)


class TestLlama70BConfig:
    def test_70b_parameter_count(self):
        """The 70B config must actually be Llama-3-70B-sized (the 8-GPU HSDP
        bench path instantiates it; meta device keeps this test cheap)."""
        from torchft_amd.models.llama import LLAMA3_70B, Llama

        with torch.device("meta"):
            m = Llama(LLAMA3_70B, dtype=torch.bfloat16,
                      checkpoint_activations=True)
        n = m.num_params()
        assert 68e9 < n < 73e9, f"70B config has {n/1e9:.1f}B params"

    def test_8b_parameter_count(self):
        from torchft_amd.models.llama import LLAMA3_8B, Llama

        with torch.device("meta"):
            m = Llama(LLAMA3_8B, dtype=torch.bfloat16)
        n = m.num_params()
        assert 7.5e9 < n < 8.5e9, f"8B config has {n/1e9:.1f}B params"


class TestFusedAdamWDTensor:
    def test_sharded_params_match_plain(self):
        """FusedAdamW on DTensor (FSDP2-sharded) params must produce the
        same values as on plain tensors (enables the HSDP fused path)."""
        import os

        import torch.distributed as dist
        from torch.distributed.device_mesh import init_device_mesh
        from torch.distributed.tensor import DTensor, Shard, distribute_tensor

        from torchft_amd.ops import FusedAdamW

        store = dist.TCPStore("127.0.0.1", 0, is_master=True,
                              wait_for_workers=False)
        dist.init_process_group("gloo", store=store, rank=0, world_size=1)
        try:
            mesh = init_device_mesh("cpu", (1,))
            torch.manual_seed(17)
            base = torch.randn(64, 8)
            p_plain = base.clone().requires_grad_(True)
            p_dt = torch.nn.Parameter(
                distribute_tensor(base.clone(), mesh, [Shard(0)])
            )
            opt_a = FusedAdamW([p_plain], lr=0.05)
            opt_b = FusedAdamW([p_dt], lr=0.05)
            for step in range(4):
                g = torch.randn(64, 8)
                p_plain.grad = g.clone()
                p_dt.grad = distribute_tensor(g.clone(), mesh, [Shard(0)])
                opt_a.step()
                opt_b.step()
            torch.testing.assert_close(
                p_dt.to_local(), p_plain.detach(), rtol=1e-5, atol=1e-6
            )
        finally:
            dist.destroy_process_group()


class TestPackedGLU:
    def test_swiglu_glu_matches_unpacked(self):
        from torchft_amd.ops import swiglu_glu, swiglu_ref

        torch.manual_seed(0)
        gu = torch.randn(3, 5, 8, dtype=torch.float32, requires_grad=True)
        out = swiglu_glu(gu)
        f = gu.shape[-1] // 2
        ref = swiglu_ref(gu.detach()[..., :f], gu.detach()[..., f:])
        torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-5)
        # backward must route gradients to both halves
        out.sum().backward()
        assert gu.grad is not None
        assert gu.grad[..., :f].abs().sum() > 0
        assert gu.grad[..., f:].abs().sum() > 0


class TestRopeTables:
    def test_shapes_and_first_position(self):
        from torchft_amd.ops import rope_tables

        cos, sin = rope_tables(seq_len=16, head_dim=8, theta=500000.0)
        assert cos.shape == (16, 4) and sin.shape == (16, 4)
        # position 0: cos=1, sin=0 for every frequency
        torch.testing.assert_close(cos[0], torch.ones(4))
        torch.testing.assert_close(sin[0], torch.zeros(4))

    def test_rope_preserves_norm(self):
        from torchft_amd.ops import rope, rope_ref, rope_tables

        torch.manual_seed(1)
        cos, sin = rope_tables(seq_len=12, head_dim=16)
        x = torch.randn(2, 12, 3, 16)
        y = rope(x, cos, sin)
        torch.testing.assert_close(y, rope_ref(x, cos, sin), rtol=1e-5, atol=1e-5)
        # a rotation: per-pair L2 norms unchanged
        torch.testing.assert_close(
            y.pow(2).sum(-1), x.pow(2).sum(-1), rtol=1e-4, atol=1e-4
        )
