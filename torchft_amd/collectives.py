"""Quantized collective algorithms over RCCL.

``allreduce_quantized`` decomposes the allreduce into
quantize → **alltoall** of per-rank slices → fused local reduce →
**allgather** of the reduced slice → dequantize, cutting wire traffic ~4x
at fp8 vs bf16 (8x vs fp32). On MI355X xGMI is 7 point-to-point links per
GPU, so the alltoall+allgather decomposition engages all links instead of
serializing on a single ring.

Reference parity (semantics): torchft/collectives.py:159-415.
"""

from __future__ import annotations

from datetime import timedelta
from typing import List, Optional

import torch
from torch.distributed import ReduceOp, Work
from torch.distributed.distributed_c10d import AllgatherOptions, AllToAllOptions
from torch.futures import Future

from torchft_amd import quantization as Q
from torchft_amd.process_group import ProcessGroup


class _QuantizedWork(Work):
    """Completes when the dequantize at the tail of the pipeline is enqueued;
    wait() makes the caller's current stream depend on the sync stream."""

    def __init__(
        self,
        fut: Future,
        sync_stream: torch.cuda.Stream,
        tensors: List[torch.Tensor],
    ) -> None:
        super().__init__()
        self._fut = fut
        self._sync_stream = sync_stream
        self._tensors = tensors
        self._event: Optional[torch.cuda.Event] = None

    def _ensure_event(self) -> None:
        if self._event is None:
            self._fut.wait()
            with torch.cuda.stream(self._sync_stream):
                self._event = torch.cuda.Event()
                self._event.record()

    def wait(self, timeout: Optional[timedelta] = None) -> bool:
        self._ensure_event()
        assert self._event is not None
        self._event.wait()  # current stream waits on the dequant tail
        return True

    def block_current_stream(self, timeout: Optional[timedelta] = None) -> None:
        self.wait()

    def get_future(self) -> Future:
        return self._fut


def allreduce_quantized(
    tensors: List[torch.Tensor],
    op: ReduceOp,
    pg: ProcessGroup,
    sync_stream: Optional[torch.cuda.Stream] = None,
) -> Work:
    """Fault-tolerant fp8 allreduce of ``tensors`` (in place).

    op must be SUM or AVG (AVG folds 1/world into the requantization scale
    inside the fused reduce kernel — no extra pass).
    """
    assert torch.cuda.is_available(), "quantized allreduce requires a HIP device"
    world = pg.size()
    avg = op == ReduceOp.AVG

    _, _, _, slice_bytes = Q.pack_geometry(tensors, world)
    device = tensors[0].device

    if sync_stream is None:
        sync_stream = torch.cuda.Stream()

    # the comm/kernel pipeline runs on sync_stream, ordered after the
    # caller's current stream
    sync_stream.wait_stream(torch.cuda.current_stream())

    with torch.cuda.stream(sync_stream):
        pack = torch.empty(world * slice_bytes, dtype=torch.uint8, device=device)
        Q.quantize_pack(tensors, pack, world)

        recv = torch.empty_like(pack)  # world copies of OUR slice
        a2a_work = pg.alltoall_base(recv, pack, [], [], AllToAllOptions())
        a2a_work.wait()

        my_slice = pack.narrow(0, 0, slice_bytes)  # reuse pack's first slice slot
        Q.reduce_slices(recv.view(world, slice_bytes).view(-1), my_slice, world, avg)

        ag_work = pg.allgather_into_tensor_coalesced([pack], [my_slice.clone()],
                                                     AllgatherOptions())
        fut = ag_work.get_future()

        def _dequant(f: Future) -> List[torch.Tensor]:
            with torch.cuda.stream(sync_stream):
                f.wait()
                Q.dequantize_pack(tensors, pack, world)
            return tensors

        fut = fut.then(_dequant)

    return _QuantizedWork(fut, sync_stream, tensors)


def reduce_scatter_quantized(
    tensors: List[torch.Tensor],
    op: ReduceOp,
    pg: ProcessGroup,
    sync_stream: Optional[torch.cuda.Stream] = None,
) -> tuple[Work, torch.Tensor]:
    """fp8 reduce-scatter: returns (work, out) where ``out`` is this rank's
    reduced share — ``blocks_per_rank * QBLOCK`` elements of the
    block-padded concatenation of ``tensors``, dequantized to the input
    dtype (zero-padded tail)."""
    assert torch.cuda.is_available(), "quantized reduce_scatter requires a HIP device"
    world = pg.size()
    avg = op == ReduceOp.AVG

    _, _, bpr, slice_bytes = Q.pack_geometry(tensors, world)
    device = tensors[0].device
    if sync_stream is None:
        sync_stream = torch.cuda.Stream()
    sync_stream.wait_stream(torch.cuda.current_stream())

    out = torch.empty(bpr * Q.QBLOCK, dtype=tensors[0].dtype, device=device)

    with torch.cuda.stream(sync_stream):
        pack = torch.empty(world * slice_bytes, dtype=torch.uint8, device=device)
        Q.quantize_pack(tensors, pack, world)

        recv = torch.empty_like(pack)
        pg.alltoall_base(recv, pack, [], [], AllToAllOptions()).wait()

        my_slice = torch.empty(slice_bytes, dtype=torch.uint8, device=device)
        Q.reduce_slices(recv, my_slice, world, avg)
        # dequantize only our slice into the flat output
        Q.dequantize_pack([out], my_slice, 1)

        fut: Future = Future()
        fut.set_result([out])

    return _QuantizedWork(fut, sync_stream, [out]), out
