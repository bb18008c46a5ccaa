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
            out = cp_attention(qs, ks, vs, pg, rank, world, causal)
            pg.shutdown()
            return out

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


class TestMergedFlashBackward:
    """The two-block LSE merge must be exact in forward AND backward
    (the backward re-enters flash with the merged out/lse)."""

    def _check(self, Hq, Hkv, dtype, atol):
        from torchft_amd.parallel.cp import _MergedFlashAttn

        torch.manual_seed(3)
        B, s, prefix, D = 2, 16, 32, 32
        S = prefix + s
        q = torch.randn(B, Hq, s, D, dtype=dtype, requires_grad=True)
        k = torch.randn(B, Hkv, S, D, dtype=dtype, requires_grad=True)
        v = torch.randn(B, Hkv, S, D, dtype=dtype, requires_grad=True)

        out = _MergedFlashAttn.apply(q, k, v, prefix)
        g = torch.randn_like(out)
        out.backward(g)
        dq, dk, dv = q.grad.clone(), k.grad.clone(), v.grad.clone()

        # oracle: full attention with the CP causal mask, plain autograd
        q2 = q.detach().clone().requires_grad_(True)
        k2 = k.detach().clone().requires_grad_(True)
        v2 = v.detach().clone().requires_grad_(True)
        import torch.nn.functional as F
        mask = torch.zeros(s, S, dtype=torch.bool)
        for i in range(s):
            mask[i, : prefix + i + 1] = True
        ref = F.scaled_dot_product_attention(
            q2, k2, v2, attn_mask=mask, enable_gqa=(Hq != Hkv)
        )
        ref.backward(g)

        torch.testing.assert_close(out, ref, rtol=1e-3, atol=atol)
        torch.testing.assert_close(dq, q2.grad, rtol=1e-3, atol=atol)
        torch.testing.assert_close(dk, k2.grad, rtol=1e-3, atol=atol)
        torch.testing.assert_close(dv, v2.grad, rtol=1e-3, atol=atol)

    def test_mha_fp32(self):
        self._check(4, 4, torch.float32, 1e-4)

    def test_gqa_fp32(self):
        self._check(4, 2, torch.float32, 1e-4)


class TestLlamaCP:
    """A CP-sharded Llama must produce the same hidden states (and loss
    gradients) as the full-sequence model."""

    def test_forward_matches_full(self):
        from torchft_amd.models.llama import LLAMA_DEBUG, CPPlan, Llama

        torch.manual_seed(7)
        full = Llama(LLAMA_DEBUG, dtype=torch.float32,
                     checkpoint_activations=False)
        full.eval()
        B, S, world = 2, 64, 2
        toks = torch.randint(0, LLAMA_DEBUG.vocab_size, (B, S))
        with torch.no_grad():
            ref = full.forward_hidden(toks)

        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
        addr = f"127.0.0.1:{store.port}/cpllama"
        sd = full.state_dict()

        def worker(rank):
            pg = ProcessGroupGloo(timeout=timedelta(seconds=20))
            pg.configure(addr, f"r{rank}", rank, world)
            m = Llama(LLAMA_DEBUG, dtype=torch.float32,
                      checkpoint_activations=False,
                      cp=CPPlan(pg=pg, rank=rank, world=world))
            m.load_state_dict(sd)
            m.eval()
            shard = shard_sequence(toks, rank, world, dim=1)
            with torch.no_grad():
                out = m.forward_hidden(shard)
            pg.shutdown()
            return out

        with ThreadPoolExecutor(max_workers=world) as ex:
            outs = list(ex.map(worker, range(world)))
        got = torch.cat(outs, dim=1)
        torch.testing.assert_close(got, ref, rtol=2e-4, atol=2e-4)


class TestRingAttention:
    """Ring-rotated KV CP: forward AND backward must match full attention."""

    def _run(self, causal, Hq, Hkv):
        from torchft_amd.parallel.cp import ring_attention

        torch.manual_seed(5)
        B, S, D, world = 2, 64, 32, 2
        q = torch.randn(B, S, Hq, D, requires_grad=True)
        k = torch.randn(B, S, Hkv, D, requires_grad=True)
        v = torch.randn(B, S, Hkv, D, requires_grad=True)
        g = torch.randn(B, S, Hq, D)
        ref = _full_attention(q, k, v, causal)
        ref.backward(g)

        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
        addr = f"127.0.0.1:{store.port}/ring{int(causal)}{Hq}"

        def worker(rank):
            pg = ProcessGroupGloo(timeout=timedelta(seconds=20))
            pg.configure(addr, f"r{rank}", rank, world)
            qs = shard_sequence(q.detach(), rank, world).requires_grad_(True)
            ks = shard_sequence(k.detach(), rank, world).requires_grad_(True)
            vs = shard_sequence(v.detach(), rank, world).requires_grad_(True)
            out = ring_attention(qs, ks, vs, pg, rank, world, causal)
            out.backward(shard_sequence(g, rank, world))
            pg.shutdown()
            return out.detach(), qs.grad, ks.grad, vs.grad

        with ThreadPoolExecutor(max_workers=world) as ex:
            results = list(ex.map(worker, range(world)))

        outs = torch.cat([r[0] for r in results], dim=1)
        dqs = torch.cat([r[1] for r in results], dim=1)
        dks = torch.cat([r[2] for r in results], dim=1)
        dvs = torch.cat([r[3] for r in results], dim=1)
        torch.testing.assert_close(outs, ref.detach(), rtol=2e-4, atol=2e-4)
        torch.testing.assert_close(dqs, q.grad, rtol=2e-4, atol=2e-4)
        torch.testing.assert_close(dks, k.grad, rtol=2e-4, atol=2e-4)
        torch.testing.assert_close(dvs, v.grad, rtol=2e-4, atol=2e-4)

    def test_causal_mha(self):
        self._run(True, 4, 4)

    def test_causal_gqa(self):
        self._run(True, 4, 2)

    def test_noncausal(self):
        self._run(False, 4, 4)


class TestZigzagRingAttention:
    """Balanced causal ring over zigzag shards: forward and backward must
    match full attention after unsharding."""

    def _run(self, Hq, Hkv, world):
        from torchft_amd.parallel.cp import (
            ring_attention_zigzag,
            shard_sequence_zigzag,
            unshard_sequence_zigzag,
        )

        torch.manual_seed(9)
        B, S, D = 2, 64, 32
        q = torch.randn(B, S, Hq, D, requires_grad=True)
        k = torch.randn(B, S, Hkv, D, requires_grad=True)
        v = torch.randn(B, S, Hkv, D, requires_grad=True)
        g = torch.randn(B, S, Hq, D)
        ref = _full_attention(q, k, v, causal=True)
        ref.backward(g)

        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
        addr = f"127.0.0.1:{store.port}/zz{Hq}{world}"

        def worker(rank):
            pg = ProcessGroupGloo(timeout=timedelta(seconds=20))
            pg.configure(addr, f"r{rank}", rank, world)
            qs = shard_sequence_zigzag(q.detach(), rank, world).requires_grad_(True)
            ks = shard_sequence_zigzag(k.detach(), rank, world).requires_grad_(True)
            vs = shard_sequence_zigzag(v.detach(), rank, world).requires_grad_(True)
            out = ring_attention_zigzag(qs, ks, vs, pg, rank, world)
            out.backward(shard_sequence_zigzag(g, rank, world))
            pg.shutdown()
            return out.detach(), qs.grad, ks.grad, vs.grad

        with ThreadPoolExecutor(max_workers=world) as ex:
            results = list(ex.map(worker, range(world)))

        outs = unshard_sequence_zigzag([r[0] for r in results], world)
        dqs = unshard_sequence_zigzag([r[1] for r in results], world)
        dks = unshard_sequence_zigzag([r[2] for r in results], world)
        dvs = unshard_sequence_zigzag([r[3] for r in results], world)
        torch.testing.assert_close(outs, ref.detach(), rtol=2e-4, atol=2e-4)
        torch.testing.assert_close(dqs, q.grad, rtol=2e-4, atol=2e-4)
        torch.testing.assert_close(dks, k.grad, rtol=2e-4, atol=2e-4)
        torch.testing.assert_close(dvs, v.grad, rtol=2e-4, atol=2e-4)

    def test_world2_mha(self):
        self._run(4, 4, 2)

    def test_world2_gqa(self):
        self._run(4, 2, 2)


class TestShardRoundtrips:
    """shard/unshard are exact inverses — property-tested over shapes."""

    def test_zigzag_roundtrip(self):
        from hypothesis import given, settings
        from hypothesis import strategies as st

        from torchft_amd.parallel.cp import (
            shard_sequence_zigzag,
            unshard_sequence_zigzag,
        )

        @settings(max_examples=25, deadline=None)
        @given(
            world=st.sampled_from([1, 2, 4]),
            chunk=st.integers(1, 8),
            batch=st.integers(1, 3),
            dim_extra=st.integers(1, 4),
        )
        def run(world, chunk, batch, dim_extra):
            seq = 2 * world * chunk
            t = torch.arange(batch * seq * dim_extra, dtype=torch.float32)
            t = t.reshape(batch, seq, dim_extra)
            parts = [shard_sequence_zigzag(t, r, world) for r in range(world)]
            # every rank owns the same amount of sequence
            assert all(p.size(1) == seq // world for p in parts)
            out = unshard_sequence_zigzag(parts, world)
            torch.testing.assert_close(out, t, rtol=0, atol=0)

        run()

    def test_zigzag_pairs_balance_causal_work(self):
        # rank r owns chunks (r, 2W-1-r): chunk index sums are equal across
        # ranks, so causal attention work per ring hop is balanced
        world = 4
        sums = [r + (2 * world - 1 - r) for r in range(world)]
        assert len(set(sums)) == 1

    def test_contiguous_shard_roundtrip(self):
        from torchft_amd.parallel.cp import shard_sequence

        t = torch.randn(2, 24, 3)
        world = 4
        parts = [shard_sequence(t, r, world) for r in range(world)]
        torch.testing.assert_close(torch.cat(parts, dim=1), t, rtol=0, atol=0)
