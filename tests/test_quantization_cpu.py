"""CPU tests of the fp8 pack format via the pure-torch reference
implementation (ground truth for the GPU kernel tests)."""

import torch

from torchft_amd.quantization import (
    QBLOCK,
    dequantize_pack_ref,
    pack_geometry,
    quantize_pack_ref,
)


class TestPackFormat:
    def test_geometry(self):
        t = [torch.zeros(5000), torch.zeros(2048)]
        total, padded, bpr, slice_bytes = pack_geometry(t, world=2)
        assert total == 3 + 1  # ceil(5000/2048)=3, 1
        assert padded == 4 and bpr == 2
        assert slice_bytes == 2 * (4 + QBLOCK)

    def test_roundtrip_ref(self):
        torch.manual_seed(0)
        tensors = [torch.randn(3000), torch.randn(2048)]
        pack = quantize_pack_ref(tensors, world=2)
        out = dequantize_pack_ref([3000, 2048], pack, world=2)
        for t, o in zip(tensors, out):
            torch.testing.assert_close(o, t, rtol=0.1, atol=0.1)

    def test_zero_block(self):
        tensors = [torch.zeros(2048)]
        pack = quantize_pack_ref(tensors, world=1)
        out = dequantize_pack_ref([2048], pack, world=1)
        assert (out[0] == 0).all()


class TestPackFormatProperties:
    """Property tests: the wire format must hold its error bound and
    geometry invariants for arbitrary tensor sizes and world sizes."""

    def test_roundtrip_residual_bound_random_shapes(self):
        from hypothesis import given, settings
        from hypothesis import strategies as st

        @settings(max_examples=30, deadline=None)
        @given(
            sizes=st.lists(st.integers(1, 5000), min_size=1, max_size=4),
            world=st.sampled_from([1, 2, 4]),
            scale_pow=st.integers(-8, 8),
            seed=st.integers(0, 2**31 - 1),
        )
        def run(sizes, world, scale_pow, seed):
            gen = torch.Generator().manual_seed(seed)
            tensors = [
                torch.randn(n, generator=gen) * (2.0 ** scale_pow) for n in sizes
            ]
            pack = quantize_pack_ref(tensors, world=world)
            out = dequantize_pack_ref(sizes, pack, world=world)
            for t, o in zip(tensors, out):
                assert o.shape == t.shape
                denom = t.norm().item() or 1.0
                # e4m3 block format: top-of-block ulp = amax/14; residual
                # norm on gaussian data stays well under 8% of tensor norm
                assert (o - t).norm().item() / denom < 0.08

        run()

    def test_geometry_invariants(self):
        from hypothesis import given, settings
        from hypothesis import strategies as st

        @settings(max_examples=50, deadline=None)
        @given(
            sizes=st.lists(st.integers(1, 10000), min_size=1, max_size=5),
            world=st.sampled_from([1, 2, 4, 8]),
        )
        def run(sizes, world):
            tensors = [torch.zeros(n) for n in sizes]
            total, padded, bpr, slice_bytes = pack_geometry(tensors, world=world)
            assert total == sum((n + QBLOCK - 1) // QBLOCK for n in sizes)
            assert padded >= total and padded % world == 0
            assert bpr == padded // world
            assert slice_bytes == bpr * (4 + QBLOCK)

        run()

    def test_outlier_block_keeps_other_blocks_precise(self):
        # block scaling is local: a huge outlier in block 0 must not
        # destroy precision in block 1 (per-tensor scaling would)
        t = torch.cat([torch.full((QBLOCK,), 1e4), torch.randn(QBLOCK)])
        pack = quantize_pack_ref([t], world=1)
        (out,) = dequantize_pack_ref([2 * QBLOCK], pack, world=1)
        tail, tail_ref = out[QBLOCK:], t[QBLOCK:]
        assert (tail - tail_ref).norm().item() / tail_ref.norm().item() < 0.08
