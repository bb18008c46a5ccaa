"""CPU validation of the quantized-allreduce choreography at world > 1.

The HIP kernels only run on MI355X, but the SLICE MATH — per-rank pack
layout, alltoall exchange, which slice each rank reduces, allgather
reassembly — is device-independent. This test runs the exact buffer
choreography of ``collectives.allreduce_quantized`` across 2 gloo ranks
using the pure-torch fp8 reference implementation in place of the kernels,
so layout/ordering bugs surface here instead of on an 8-GPU node.
"""

import struct
from concurrent.futures import ThreadPoolExecutor
from datetime import timedelta
from typing import List

import torch
from torch.distributed import TCPStore
from torch.distributed.distributed_c10d import AllgatherOptions, AllToAllOptions

from torchft_amd.process_group import ProcessGroupGloo
from torchft_amd.quantization import (
    FP8_MAX,
    QBLOCK,
    dequantize_pack_ref,
    pack_geometry,
    quantize_pack_ref,
)


def _reduce_slices_ref(recv: torch.Tensor, world: int, slice_bytes: int,
                       avg: bool) -> torch.Tensor:
    """Reference of the fused reduce kernel: sum W quantized copies of this
    rank's slice in fixed rank order, requantize."""
    bpr = slice_bytes // (4 + QBLOCK)
    out = torch.zeros(slice_bytes, dtype=torch.uint8)
    for bl in range(bpr):
        acc = torch.zeros(QBLOCK)
        for r in range(world):
            sl = recv[r * slice_bytes : (r + 1) * slice_bytes]
            dq = struct.unpack("<f", bytes(sl[bl * 4 : bl * 4 + 4].tolist()))[0]
            payload = sl[bpr * 4 + bl * QBLOCK : bpr * 4 + (bl + 1) * QBLOCK]
            acc += payload.view(torch.float8_e4m3fn).float() * dq
        amax = acc.abs().max()
        dq_out = float(amax / FP8_MAX) if amax > 0 else 0.0
        if avg:
            dq_out /= world
        q = (acc * (FP8_MAX / amax)).to(torch.float8_e4m3fn) if amax > 0 else \
            torch.zeros(QBLOCK, dtype=torch.float8_e4m3fn)
        out[bl * 4 : bl * 4 + 4] = torch.tensor(
            list(struct.pack("<f", dq_out)), dtype=torch.uint8
        )
        out[bpr * 4 + bl * QBLOCK : bpr * 4 + (bl + 1) * QBLOCK] = q.view(torch.uint8)
    return out


class TestQuantizedAllreduceChoreography:
    def test_world2_sum(self):
        world = 2
        torch.manual_seed(0)
        # per-rank inputs (shapes exercise per-tensor block padding)
        inputs = {
            r: [torch.randn(3000), torch.randn(2048)] for r in range(world)
        }
        expected = [
            sum(inputs[r][i] for r in range(world)) for i in range(2)
        ]

        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
        addr = f"127.0.0.1:{store.port}/qar"

        def run(rank: int) -> List[torch.Tensor]:
            tensors = [t.clone() for t in inputs[rank]]
            pg = ProcessGroupGloo(timeout=timedelta(seconds=20))
            pg.configure(addr, f"r{rank}", rank, world)
            _, _, _, slice_bytes = pack_geometry(tensors, world)

            # 1. quantize into the packed wire buffer (ref impl)
            pack = quantize_pack_ref(tensors, world)
            # 2. alltoall: rank r receives every rank's copy of slice r
            recv = torch.empty_like(pack)
            pg.alltoall_base(recv, pack, [], [], AllToAllOptions()).wait()
            # 3. fused reduce of OUR slice in fixed rank order
            my_slice = _reduce_slices_ref(recv, world, slice_bytes, avg=False)
            # 4. allgather reduced slices back into the full pack
            pg.allgather_into_tensor_coalesced(
                [pack.view(-1)], [my_slice], AllgatherOptions()
            ).wait()
            # 5. dequantize
            return dequantize_pack_ref([t.numel() for t in tensors], pack, world)

        with ThreadPoolExecutor(max_workers=world) as ex:
            results = list(ex.map(run, range(world)))

        for rank_result in results:
            for got, exp in zip(rank_result, expected):
                # two quantization passes => looser tolerance
                torch.testing.assert_close(got, exp, rtol=0.2, atol=0.2)
        # both ranks must hold the IDENTICAL reduced result (bitwise): the
        # fixed-order reduce makes rank 0's and rank 1's outputs equal
        for a, b in zip(results[0], results[1]):
            assert torch.equal(a, b), "ranks diverged after quantized allreduce"

    def test_world4_avg(self):
        # AVG folds 1/world into the requantization scale; validate the
        # same choreography at world 4 with residual-norm bounds
        world = 4
        torch.manual_seed(1)
        inputs = {
            r: [torch.randn(5000), torch.randn(100)] for r in range(world)
        }
        expected = [
            sum(inputs[r][i] for r in range(world)) / world for i in range(2)
        ]

        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
        addr = f"127.0.0.1:{store.port}/qar4"

        def run(rank: int) -> List[torch.Tensor]:
            tensors = [t.clone() for t in inputs[rank]]
            pg = ProcessGroupGloo(timeout=timedelta(seconds=20))
            try:
                pg.configure(addr, f"r{rank}", rank, world)
                _, _, _, slice_bytes = pack_geometry(tensors, world)
                pack = quantize_pack_ref(tensors, world)
                recv = torch.empty_like(pack)
                pg.alltoall_base(recv, pack, [], [], AllToAllOptions()).wait()
                my_slice = _reduce_slices_ref(recv, world, slice_bytes, avg=True)
                pg.allgather_into_tensor_coalesced(
                    [pack.view(-1)], [my_slice], AllgatherOptions()
                ).wait()
                return dequantize_pack_ref([t.numel() for t in tensors], pack, world)
            finally:
                pg.shutdown()

        with ThreadPoolExecutor(max_workers=world) as ex:
            results = list(ex.map(run, range(world)))

        for rank_result in results:
            for got, exp in zip(rank_result, expected):
                # residual-norm bound: pointwise tolerances are the wrong
                # shape for block-scaled fp8 (top-of-block ulp = amax/14)
                assert (got - exp).norm() / exp.norm() < 0.08
        for r in range(1, world):
            for a, b in zip(results[0], results[r]):
                assert torch.equal(a, b), "ranks diverged after quantized allreduce"
