"""GPU numerics tests: every CDNA4 HIP kernel vs a plain PyTorch fp32
reference of the same op (run on a real MI355X via `pytest -m gpu`)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def setup_module(module):
    if torch.cuda.is_available():
        torch.cuda.init()


@pytest.fixture
def dev():
    return torch.device("cuda:0")


class TestRMSNorm:
    @pytest.mark.parametrize("rows,H", [(128, 4096), (64, 8192), (33, 256)])
    def test_forward(self, dev, rows, H):
        from torchft_amd.ops import hip_ext, rmsnorm_ref

        torch.manual_seed(0)
        x = torch.randn(rows, H, device=dev, dtype=torch.bfloat16)
        w = torch.randn(H, device=dev, dtype=torch.bfloat16)
        y, invrms = hip_ext().rmsnorm_fwd(x, w, 1e-5)
        ref = rmsnorm_ref(x, w, 1e-5)
        torch.testing.assert_close(y, ref, rtol=1e-2, atol=1e-2)

    def test_backward(self, dev):
        from torchft_amd.ops import rmsnorm

        torch.manual_seed(1)
        rows, H = 96, 4096
        x = torch.randn(rows, H, device=dev, dtype=torch.bfloat16, requires_grad=True)
        w = torch.randn(H, device=dev, dtype=torch.bfloat16, requires_grad=True)
        y = rmsnorm(x, w, 1e-5)
        dy = torch.randn_like(y)
        y.backward(dy)

        xf = x.detach().float().requires_grad_(True)
        wf = w.detach().float().requires_grad_(True)
        inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-5)
        (xf * inv * wf).backward(dy.float())

        torch.testing.assert_close(x.grad.float(), xf.grad, rtol=5e-2, atol=5e-2)
        torch.testing.assert_close(w.grad.float(), wf.grad, rtol=5e-2, atol=5e-1)


class TestRope:
    def test_forward_matches_ref(self, dev):
        from torchft_amd.ops import hip_ext, rope_ref, rope_tables

        torch.manual_seed(2)
        B, S, Hh, D = 2, 64, 4, 128
        x = torch.randn(B, S, Hh, D, device=dev, dtype=torch.bfloat16)
        cos, sin = rope_tables(S, D, device=dev)
        out = hip_ext().rope_apply(x, cos, sin, False)
        ref = rope_ref(x, cos, sin)
        torch.testing.assert_close(out, ref, rtol=2e-2, atol=2e-2)

    def test_backward_is_transpose(self, dev):
        from torchft_amd.ops import rope, rope_tables

        torch.manual_seed(3)
        B, S, Hh, D = 1, 32, 2, 64
        cos, sin = rope_tables(S, D, device=dev)
        x = torch.randn(B, S, Hh, D, device=dev, dtype=torch.bfloat16, requires_grad=True)
        out = rope(x, cos, sin)
        dy = torch.randn_like(out)
        out.backward(dy)

        xf = x.detach().float().requires_grad_(True)
        c = cos[:S].view(1, S, 1, D // 2)
        s = sin[:S].view(1, S, 1, D // 2)
        x1, x2 = xf.chunk(2, dim=-1)
        ref = torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1)
        ref.backward(dy.float())
        torch.testing.assert_close(x.grad.float(), xf.grad, rtol=2e-2, atol=2e-2)


class TestSwiGLU:
    def test_fwd_bwd(self, dev):
        from torchft_amd.ops import swiglu

        torch.manual_seed(4)
        a = torch.randn(4096 * 4, device=dev, dtype=torch.bfloat16, requires_grad=True)
        b = torch.randn_like(a, requires_grad=True)
        out = swiglu(a, b)
        dy = torch.randn_like(out)
        out.backward(dy)

        af = a.detach().float().requires_grad_(True)
        bf = b.detach().float().requires_grad_(True)
        ref = torch.nn.functional.silu(af) * bf
        ref.backward(dy.float())
        torch.testing.assert_close(out.float(), ref.detach(), rtol=2e-2, atol=2e-2)
        torch.testing.assert_close(a.grad.float(), af.grad, rtol=3e-2, atol=3e-2)
        torch.testing.assert_close(b.grad.float(), bf.grad, rtol=3e-2, atol=3e-2)


class TestFp8Quant:
    @pytest.mark.parametrize("world", [1, 2, 4])
    def test_quantize_dequantize_roundtrip(self, dev, world):
        from torchft_amd.ops import hip_ext
        from torchft_amd.quantization import allocate_pack

        torch.manual_seed(5)
        tensors = [
            torch.randn(3000, device=dev, dtype=torch.bfloat16),
            torch.randn(17, 129, device=dev, dtype=torch.bfloat16),
            torch.randn(2048, device=dev, dtype=torch.bfloat16),
        ]
        orig = [t.clone() for t in tensors]
        pack = allocate_pack(tensors, world)
        hip_ext().fp8_quantize(tensors, pack, world)
        for t in tensors:
            t.zero_()
        hip_ext().fp8_dequantize(tensors, pack, world)
        for t, o in zip(tensors, orig):
            # fp8 e4m3 block quantization: ~2 decimal digits
            torch.testing.assert_close(t.float(), o.float(), rtol=0.1, atol=0.1)

    def test_pack_matches_torch_reference(self, dev):
        from torchft_amd.ops import hip_ext
        from torchft_amd.quantization import allocate_pack, quantize_pack_ref

        torch.manual_seed(6)
        tensors = [torch.randn(2048 * 2 + 100, device=dev, dtype=torch.float32)]
        pack = allocate_pack(tensors, 2)
        hip_ext().fp8_quantize(tensors, pack, 2)
        ref = quantize_pack_ref(tensors, 2)
        # payload bytes should match the torch float8 cast bit-for-bit
        # (both round-to-nearest-even into OCP e4m3)
        assert (pack == ref).float().mean().item() > 0.99

    def test_reduce_sums_copies(self, dev):
        from torchft_amd.ops import hip_ext
        from torchft_amd.quantization import allocate_pack, pack_geometry

        torch.manual_seed(7)
        world = 2
        t = [torch.randn(4096, device=dev, dtype=torch.float32)]
        _, _, bpr, slice_bytes = pack_geometry(t, world)

        # two ranks' worth of quantized data for the same logical tensors
        packs = []
        vals = []
        for r in range(world):
            v = [torch.randn(4096, device=dev, dtype=torch.float32)]
            vals.append(v[0].clone())
            p = allocate_pack(v, world)
            hip_ext().fp8_quantize(v, p, world)
            packs.append(p)

        # emulate the alltoall result for slice 0: both ranks' slice 0
        recv = torch.cat([p[:slice_bytes] for p in packs])
        out = torch.empty(slice_bytes, dtype=torch.uint8, device=dev)
        hip_ext().fp8_reduce(recv, out, world, False)

        # decode: dequantize out (one slice covering blocks 0..bpr)
        dec = [torch.zeros(min(4096, bpr * 2048), device=dev, dtype=torch.float32)]
        hip_ext().fp8_dequantize(dec, out, 1)
        expected = (vals[0] + vals[1])[: dec[0].numel()]
        torch.testing.assert_close(dec[0], expected, rtol=0.15, atol=0.15)


class TestFusedAdamW:
    def test_matches_torch_adamw(self, dev):
        from torchft_amd.ops import FusedAdamW

        torch.manual_seed(8)
        p1 = torch.randn(5000, device=dev, dtype=torch.bfloat16, requires_grad=True)
        p2 = torch.randn(100, 33, device=dev, dtype=torch.bfloat16, requires_grad=True)
        ref1 = p1.detach().float().clone().requires_grad_(True)
        ref2 = p2.detach().float().clone().requires_grad_(True)

        opt = FusedAdamW([p1, p2], lr=1e-2, betas=(0.9, 0.95), weight_decay=0.01)
        ref_opt = torch.optim.AdamW(
            [ref1, ref2], lr=1e-2, betas=(0.9, 0.95), weight_decay=0.01, eps=1e-8
        )
        for step in range(5):
            g1 = torch.randn_like(p1)
            g2 = torch.randn_like(p2)
            p1.grad = g1
            p2.grad = g2
            ref1.grad = g1.float()
            ref2.grad = g2.float()
            opt.step()
            ref_opt.step()
        torch.testing.assert_close(p1.float(), ref1.detach(), rtol=2e-2, atol=2e-2)
        torch.testing.assert_close(p2.float(), ref2.detach(), rtol=2e-2, atol=2e-2)


class TestQuantizedAllreduce:
    def test_world1_identity_up_to_quantization(self, dev):
        """allreduce_quantized on a world-1 RCCL pg: the full
        quantize→alltoall→reduce→allgather→dequantize pipeline must be the
        identity up to fp8 error."""
        import os

        from torch.distributed import TCPStore

        from torchft_amd.collectives import allreduce_quantized
        from torchft_amd.process_group import ProcessGroupRCCL
        from torch.distributed import ReduceOp

        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
        pg = ProcessGroupRCCL()
        pg.configure(f"127.0.0.1:{store.port}/q", "r0", 0, 1)
        torch.manual_seed(9)
        t = torch.randn(100_000, device=dev, dtype=torch.bfloat16)
        orig = t.clone()
        work = allreduce_quantized([t], ReduceOp.SUM, pg)
        work.wait()
        torch.cuda.synchronize()
        torch.testing.assert_close(t.float(), orig.float(), rtol=0.1, atol=0.1)
        pg.shutdown()


class TestLlamaSmokeGPU:
    def test_tiny_forward_backward(self, dev):
        from torchft_amd.models.llama import Llama, LlamaConfig
        from torchft_amd.ops import FusedAdamW

        cfg = LlamaConfig(
            dim=512, n_layers=2, n_heads=8, n_kv_heads=4, ffn_hidden=1024,
            vocab_size=1024, max_seq_len=256,
        )
        torch.manual_seed(10)
        model = Llama(cfg, dtype=torch.bfloat16, checkpoint_activations=False).to(dev)
        opt = FusedAdamW(model.parameters(), lr=1e-4)
        toks = torch.randint(0, cfg.vocab_size, (2, 129), device=dev)
        loss = model.forward_loss(toks[:, :-1], toks[:, 1:])
        loss.backward()
        opt.step()
        torch.cuda.synchronize()
        assert torch.isfinite(loss)



def _assert_wire_close(got: torch.Tensor, exact: torch.Tensor, hops: int,
                       msg: str) -> None:
    """fp8 e4m3's ulp near the top of a 2048-block is amax/14 (~0.28 for
    N(0,1) inputs; measured: scripts/debug_quant_pipeline.py), so pointwise
    tolerances cannot separate wire noise from a broken exchange. Assert
    the residual NORM instead: quantization noise keeps it ~1-2% of the
    signal; any slice/ordering bug pushes it to O(1).
    """
    got = got.float()
    exact = exact.float()
    resid = (got - exact).norm() / exact.norm().clamp_min(1e-12)
    assert resid < 0.05, f"{msg}: residual norm {resid:.4f}"
    max_abs = (got - exact).abs().max().item()
    assert max_abs < 0.35 * hops * max(1.0, exact.abs().max().item() / 3.5), (
        f"{msg}: max abs {max_abs:.3f}"
    )


class _MailboxCtx:
    """Shared exchange state for N virtual ranks on one device."""

    def __init__(self, world: int):
        import threading

        self.world = world
        self.barrier = threading.Barrier(world)
        self.lock = threading.Lock()
        self.posts = {}

    def post(self, key, rank, val):
        with self.lock:
            self.posts.setdefault(key, {})[rank] = val

    def get(self, key, rank):
        with self.lock:
            return self.posts[key][rank]


class _MailboxPG:
    """World-N process-group double on ONE GPU: N thread-ranks exchange
    device buffers through a mailbox with event-ordered copies. Exercises
    the real alltoall/allgather data movement of the quantized collectives
    without needing N devices (RCCL cannot place two ranks on one GPU)."""

    def __init__(self, ctx: _MailboxCtx, rank: int):
        self._ctx = ctx
        self._rank = rank
        self._seq = 0

    def size(self):
        return self._ctx.world

    def _exchange(self, kind, out, inp, per_rank_out, per_rank_in):
        from torchft_amd.work import _DummyWork

        ev = torch.cuda.Event()
        ev.record()  # inp was produced on the caller's current stream
        key = (kind, self._seq)
        self._seq += 1
        self._ctx.post(key, self._rank, (inp, ev))
        self._ctx.barrier.wait()
        cur = torch.cuda.current_stream()
        for src in range(self._ctx.world):
            sbuf, sev = self._ctx.get(key, src)
            cur.wait_event(sev)
            out[src * per_rank_out : (src + 1) * per_rank_out].copy_(
                sbuf[self._rank * per_rank_in : (self._rank + 1) * per_rank_in]
                if per_rank_in
                else sbuf
            )
        # publish "my copies are enqueued" and make the producer's stream
        # wait for EVERY consumer before it may overwrite its send buffer
        # (the CPU barrier alone does not order the device work)
        done = torch.cuda.Event()
        done.record()
        self._ctx.post((kind, "done", key[1]), self._rank, done)
        self._ctx.barrier.wait()
        for peer in range(self._ctx.world):
            cur.wait_event(self._ctx.get((kind, "done", key[1]), peer))
        return _DummyWork([out])

    def alltoall_base(self, out, inp, _os, _is, _opts):
        n = inp.numel() // self._ctx.world
        return self._exchange("a2a", out, inp, n, n)

    def allgather_into_tensor_coalesced(self, outs, ins, _opts):
        out, inp = outs[0], ins[0]
        return self._exchange("ag", out, inp, inp.numel(), 0)


class TestQuantizedAllreduceMultiRank:
    """Full quantize->alltoall->reduce->allgather->dequantize choreography
    with real CDNA4 kernels at world 2 and 4, thread-ranks on one MI355X
    (reference semantics torchft/collectives.py:297-415)."""

    @pytest.mark.parametrize("world", [2, 4])
    @pytest.mark.parametrize("op_avg", [False, True])
    def test_allreduce_quantized_world_n(self, dev, world, op_avg):
        from concurrent.futures import ThreadPoolExecutor

        from torch.distributed import ReduceOp

        from torchft_amd.collectives import allreduce_quantized

        torch.manual_seed(11)
        n = 100_000
        inputs = [
            torch.randn(n, device=dev, dtype=torch.bfloat16) for _ in range(world)
        ]
        expected = torch.stack([t.float() for t in inputs]).sum(0)
        if op_avg:
            expected /= world
        ctx = _MailboxCtx(world)

        def rank_main(r):
            t = inputs[r].clone()
            pg = _MailboxPG(ctx, r)
            work = allreduce_quantized(
                [t], ReduceOp.AVG if op_avg else ReduceOp.SUM, pg
            )
            work.wait()
            torch.cuda.synchronize()
            return t

        with ThreadPoolExecutor(max_workers=world) as ex:
            outs = list(ex.map(rank_main, range(world)))

        for r, out in enumerate(outs):
            _assert_wire_close(out, expected, hops=2,
                               msg=f"rank {r} at world {world}")

    @pytest.mark.parametrize("world", [2, 4])
    def test_reduce_scatter_quantized_world_n(self, dev, world):
        from concurrent.futures import ThreadPoolExecutor

        from torch.distributed import ReduceOp

        from torchft_amd import quantization as Q
        from torchft_amd.collectives import reduce_scatter_quantized

        torch.manual_seed(12)
        n = 65_536
        inputs = [
            torch.randn(n, device=dev, dtype=torch.bfloat16) for _ in range(world)
        ]
        total = torch.stack([t.float() for t in inputs]).sum(0)
        _, _, bpr, _ = Q.pack_geometry([inputs[0]], world)
        ctx = _MailboxCtx(world)

        def rank_main(r):
            t = inputs[r].clone()
            pg = _MailboxPG(ctx, r)
            work, out = reduce_scatter_quantized([t], ReduceOp.SUM, pg)
            work.wait()
            torch.cuda.synchronize()
            return out

        with ThreadPoolExecutor(max_workers=world) as ex:
            outs = list(ex.map(rank_main, range(world)))

        chunk = bpr * Q.QBLOCK
        for r, out in enumerate(outs):
            lo = r * chunk
            hi = min(lo + chunk, n)
            if lo >= n:
                continue
            _assert_wire_close(out[: hi - lo], total[lo:hi], hops=2,
                               msg=f"rank {r} slice at world {world}")


class TestSwiGLUGlu:
    def test_packed_glu_matches_ref(self, dev):
        from torchft_amd.ops import hip_ext, swiglu_ref

        torch.manual_seed(13)
        gu = torch.randn(64, 2 * 1024, device=dev, dtype=torch.bfloat16)
        out = hip_ext().swiglu_glu_fwd(gu)
        ref = swiglu_ref(gu[..., :1024].contiguous(), gu[..., 1024:].contiguous())
        torch.testing.assert_close(out, ref, rtol=2e-2, atol=2e-2)

    def test_packed_glu_backward(self, dev):
        from torchft_amd.ops import swiglu_glu

        torch.manual_seed(14)
        gu = torch.randn(32, 2 * 512, device=dev, dtype=torch.bfloat16,
                         requires_grad=True)
        y = swiglu_glu(gu)
        dy = torch.randn_like(y)
        y.backward(dy)

        guf = gu.detach().float().requires_grad_(True)
        import torch.nn.functional as F
        (F.silu(guf[..., :512]) * guf[..., 512:]).backward(dy.float())
        torch.testing.assert_close(gu.grad.float(), guf.grad, rtol=3e-2, atol=3e-2)
