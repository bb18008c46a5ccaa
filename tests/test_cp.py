"""Context-parallel attention tests: CP output must match full attention."""

from concurrent.futures import ThreadPoolExecutor
from datetime import timedelta

import torch
from torch.distributed import TCPStore

from torchft_amd.parallel.cp import cp_attention, shard_sequence, _sdpa_with_lse
from torchft_amd.process_group import ProcessGroupGloo


def _full_attention(q, k, v, causal=True):
    import torch.nn.functional as F
    out = F.scaled_dot_product_attention(
        q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
        is_causal=causal, enable_gqa=True,
    )
    return out.transpose(1, 2)


class TestSdpaWithLse:
    def test_matches_sdpa(self):
        torch.manual_seed(0)
        q = torch.randn(1, 4, 16, 32)
        k = torch.randn(1, 2, 16, 32)
        v = torch.randn(1, 2, 16, 32)
        out, lse = _sdpa_with_lse(q, k, v, causal=True)
        import torch.nn.functional as F
        ref = F.scaled_dot_product_attention(q, k, v, is_causal=True, enable_gqa=True)
        torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-4)


class TestContextParallel:
    def _run(self, world, causal):
        torch.manual_seed(1)
        B, S, H, Hkv, D = 2, 32, 4, 2, 16
        q = torch.randn(B, S, H, D)
        k = torch.randn(B, S, Hkv, D)
        v = torch.randn(B, S, Hkv, D)
        ref = _full_attention(q, k, v, causal)

        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
        addr = f"127.0.0.1:{store.port}/cp"

        def worker(rank):
            pg = ProcessGroupGloo(timeout=timedelta(seconds=20))
            pg.configure(addr, f"r{rank}", rank, world)
            qs = shard_sequence(q, rank, world)
            ks = shard_sequence(k, rank, world)
            vs = shard_sequence(v, rank, world)
            return cp_attention(qs, ks, vs, pg, rank, world, causal)

        with ThreadPoolExecutor(max_workers=world) as ex:
            outs = list(ex.map(worker, range(world)))
        got = torch.cat(outs, dim=1)
        torch.testing.assert_close(got, ref, rtol=2e-3, atol=2e-3)

    def test_causal_world2(self):
        self._run(2, True)

    def test_causal_world4(self):
        self._run(4, True)

    def test_noncausal_world2(self):
        self._run(2, False)

    def test_backward_grads_match(self):
        torch.manual_seed(2)
        B, S, H, Hkv, D = 1, 16, 2, 1, 8
        world = 2
        q = torch.randn(B, S, H, D, requires_grad=True)
        k = torch.randn(B, S, Hkv, D, requires_grad=True)
        v = torch.randn(B, S, Hkv, D, requires_grad=True)
        dy = torch.randn(B, S, H, D)
        ref = _full_attention(q, k, v, True)
        ref.backward(dy)
        ref_grads = (q.grad.clone(), k.grad.clone(), v.grad.clone())

        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
        addr = f"127.0.0.1:{store.port}/cpb"

        def worker(rank):
            pg = ProcessGroupGloo(timeout=timedelta(seconds=20))
            pg.configure(addr, f"r{rank}", rank, world)
            qs = shard_sequence(q.detach(), rank, world).requires_grad_(True)
            ks = shard_sequence(k.detach(), rank, world).requires_grad_(True)
            vs = shard_sequence(v.detach(), rank, world).requires_grad_(True)
            out = cp_attention(qs, ks, vs, pg, rank, world, True)
            out.backward(shard_sequence(dy, rank, world))
            return qs.grad, ks.grad, vs.grad

        with ThreadPoolExecutor(max_workers=world) as ex:
            grads = list(ex.map(worker, range(world)))
        dq = torch.cat([g[0] for g in grads], dim=1)
        dk = torch.cat([g[1] for g in grads], dim=1)
        dv = torch.cat([g[2] for g in grads], dim=1)
        torch.testing.assert_close(dq, ref_grads[0], rtol=2e-3, atol=2e-3)
        torch.testing.assert_close(dk, ref_grads[1], rtol=2e-3, atol=2e-3)
        torch.testing.assert_close(dv, ref_grads[2], rtol=2e-3, atol=2e-3)


class TestShardGuards:
    def test_indivisible_sequence_rejected(self):
        import pytest as _pytest

        with _pytest.raises(AssertionError):
            shard_sequence(torch.zeros(1, 30, 2, 4), 0, 4)
