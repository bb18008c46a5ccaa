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
