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
