"""Custom flash-attention backward: numerics vs stock SDPA autograd + the
MFMA layout probe (GPU)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


class TestMfmaProbe:
    def test_layout_assumptions(self):
        from torchft_amd.ops import hip_ext

        torch.manual_seed(0)
        # asymmetric operands catch transposed layouts (guide rule G9)
        A = (torch.arange(32 * 16, device="cuda").reshape(32, 16).float() % 13 - 6)
        B = (torch.arange(16 * 32, device="cuda").reshape(16, 32).float() % 7 - 3)
        A16, B16 = A.to(torch.bfloat16), B.to(torch.bfloat16)
        D = hip_ext().mfma_probe(A16, B16)
        ref = A16.float() @ B16.float()
        torch.testing.assert_close(D, ref, rtol=1e-3, atol=1e-3)


def _ref_grads(q, k, v, dout, causal):
    qf = q.detach().clone().requires_grad_(True)
    kf = k.detach().clone().requires_grad_(True)
    vf = v.detach().clone().requires_grad_(True)
    out = torch.nn.functional.scaled_dot_product_attention(
        qf, kf, vf, is_causal=causal, enable_gqa=qf.shape[1] != kf.shape[1]
    )
    out.backward(dout)
    return out.detach(), qf.grad, kf.grad, vf.grad


class TestFlashBwd:
    @pytest.mark.parametrize(
        "B,Hq,Hkv,S,causal",
        [
            (1, 2, 2, 128, True),
            (1, 4, 2, 256, True),
            (2, 4, 1, 384, True),
            (1, 2, 2, 128, False),
            (1, 8, 2, 512, True),
        ],
    )
    def test_matches_sdpa_autograd(self, B, Hq, Hkv, S, causal):
        from torchft_amd.ops.flash_attention import _FlashAttentionFn

        torch.manual_seed(1)
        dev = "cuda"
        q = torch.randn(B, Hq, S, 128, device=dev, dtype=torch.bfloat16,
                        requires_grad=True)
        k = torch.randn(B, Hkv, S, 128, device=dev, dtype=torch.bfloat16,
                        requires_grad=True)
        v = torch.randn(B, Hkv, S, 128, device=dev, dtype=torch.bfloat16,
                        requires_grad=True)
        dout = torch.randn(B, Hq, S, 128, device=dev, dtype=torch.bfloat16)

        out = _FlashAttentionFn.apply(q, k, v, causal, 128 ** -0.5)
        out.backward(dout)
        torch.cuda.synchronize()

        ref_out, ref_dq, ref_dk, ref_dv = _ref_grads(q, k, v, dout, causal)
        torch.testing.assert_close(out, ref_out, rtol=2e-2, atol=2e-2)
        torch.testing.assert_close(q.grad, ref_dq, rtol=5e-2, atol=5e-2)
        torch.testing.assert_close(k.grad, ref_dk, rtol=5e-2, atol=5e-2)
        torch.testing.assert_close(v.grad, ref_dv, rtol=5e-2, atol=5e-2)


class TestCustomForward:
    """Hand-written forward: out AND lse must match aten's flash (the
    in-tree backward consumes the lse)."""

    @pytest.mark.parametrize("causal", [True, False])
    @pytest.mark.parametrize("B,Hq,Hkv,S", [(1, 4, 2, 512), (2, 8, 8, 1024)])
    def test_fwd_matches_aten(self, causal, B, Hq, Hkv, S):
        from torchft_amd.ops import hip_ext

        torch.manual_seed(21)
        D = 128
        dev = "cuda"
        q = torch.randn(B, Hq, S, D, device=dev, dtype=torch.bfloat16)
        k = torch.randn(B, Hkv, S, D, device=dev, dtype=torch.bfloat16)
        v = torch.randn(B, Hkv, S, D, device=dev, dtype=torch.bfloat16)
        scale = D ** -0.5
        out, lse = hip_ext().fa_fwd(q, k, v, scale, causal)
        ref, ref_lse, *_ = torch.ops.aten._scaled_dot_product_flash_attention(
            q, k, v, 0.0, causal, False, scale=scale
        )
        torch.testing.assert_close(out, ref, rtol=2e-2, atol=2e-2)
        torch.testing.assert_close(lse, ref_lse.float(), rtol=1e-3, atol=1e-3)

    def test_full_custom_fwd_bwd(self, monkeypatch):
        """Custom fwd feeding the custom bwd must match stock end to end."""
        from torchft_amd.ops.flash_attention import _FlashAttentionFn

        monkeypatch.setenv("TORCHFT_AMD_CUSTOM_FA_FWD", "1")
        torch.manual_seed(22)
        B, Hq, Hkv, S, D = 1, 8, 2, 512, 128
        dev = "cuda"
        q = torch.randn(B, Hq, S, D, device=dev, dtype=torch.bfloat16,
                        requires_grad=True)
        k = torch.randn(B, Hkv, S, D, device=dev, dtype=torch.bfloat16,
                        requires_grad=True)
        v = torch.randn(B, Hkv, S, D, device=dev, dtype=torch.bfloat16,
                        requires_grad=True)
        g = torch.randn(B, Hq, S, D, device=dev, dtype=torch.bfloat16)
        out = _FlashAttentionFn.apply(q, k, v, True, D ** -0.5)
        out.backward(g)

        q2 = q.detach().clone().requires_grad_(True)
        k2 = k.detach().clone().requires_grad_(True)
        v2 = v.detach().clone().requires_grad_(True)
        import torch.nn.functional as F
        ref = F.scaled_dot_product_attention(q2, k2, v2, is_causal=True,
                                             enable_gqa=True)
        ref.backward(g)
        torch.testing.assert_close(out, ref, rtol=2e-2, atol=2e-2)
        torch.testing.assert_close(q.grad, q2.grad, rtol=5e-2, atol=5e-2)
        torch.testing.assert_close(k.grad, k2.grad, rtol=5e-2, atol=5e-2)
        torch.testing.assert_close(v.grad, v2.grad, rtol=5e-2, atol=5e-2)


class TestFaDelta:
    def test_delta_matches_torch(self):
        from torchft_amd.ops import hip_ext

        torch.manual_seed(23)
        d = torch.randn(2, 4, 256, 128, device="cuda", dtype=torch.bfloat16)
        o = torch.randn_like(d)
        got = hip_ext().fa_delta(d, o)
        ref = (d.float() * o.float()).sum(-1)
        torch.testing.assert_close(got, ref, rtol=2e-2, atol=2e-2)


class TestBshdLayout:
    """The kernels must consume [B,S,H,D]-backed transpose views (the
    model's projection layout) without any copy, matching the packed path."""

    def _mk(self, B, Hq, Hkv, S, D=128):
        torch.manual_seed(31)
        q = torch.randn(B, S, Hq, D, device="cuda", dtype=torch.bfloat16)
        k = torch.randn(B, S, Hkv, D, device="cuda", dtype=torch.bfloat16)
        v = torch.randn(B, S, Hkv, D, device="cuda", dtype=torch.bfloat16)
        return q, k, v

    def test_fwd_bshd_matches_packed(self):
        from torchft_amd.ops import hip_ext

        q, k, v = self._mk(2, 8, 4, 512)
        scale = 128 ** -0.5
        qv, kv_, vv = (t.transpose(1, 2) for t in (q, k, v))  # BSHD views
        out_v, lse_v = hip_ext().fa_fwd(qv, kv_, vv, scale, True)
        out_c, lse_c = hip_ext().fa_fwd(
            qv.contiguous(), kv_.contiguous(), vv.contiguous(), scale, True
        )
        torch.testing.assert_close(out_v, out_c, rtol=0, atol=0)
        torch.testing.assert_close(lse_v, lse_c, rtol=0, atol=0)

    def test_full_chain_bshd_no_copies(self, monkeypatch):
        """flash_attention on transpose views must match stock SDPA, and the
        gradients must come back in the projection layout."""
        from torchft_amd.ops.flash_attention import flash_attention

        monkeypatch.setenv("TORCHFT_AMD_CUSTOM_FA", "1")
        q, k, v = self._mk(1, 8, 2, 512)
        q = q.requires_grad_(True)
        k = k.requires_grad_(True)
        v = v.requires_grad_(True)
        g = torch.randn(1, 8, 512, 128, device="cuda", dtype=torch.bfloat16)

        out = flash_attention(q.transpose(1, 2), k.transpose(1, 2),
                              v.transpose(1, 2), causal=True)
        out.backward(g)

        q2 = q.detach().clone().requires_grad_(True)
        k2 = k.detach().clone().requires_grad_(True)
        v2 = v.detach().clone().requires_grad_(True)
        import torch.nn.functional as F
        ref = F.scaled_dot_product_attention(
            q2.transpose(1, 2), k2.transpose(1, 2), v2.transpose(1, 2),
            is_causal=True, enable_gqa=True,
        )
        ref.backward(g)
        torch.testing.assert_close(out, ref, rtol=2e-2, atol=2e-2)
        torch.testing.assert_close(q.grad, q2.grad, rtol=5e-2, atol=5e-2)
        torch.testing.assert_close(k.grad, k2.grad, rtol=5e-2, atol=5e-2)
        torch.testing.assert_close(v.grad, v2.grad, rtol=5e-2, atol=5e-2)
