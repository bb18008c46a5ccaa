"""Checkpoint-transport benchmarks: timed send/recv of a large synthetic
state dict (default 12 GB in line with the reference benches:
torchft/checkpointing/pg_transport_bench.py, http_transport_bench.py).

    python -m torchft_amd.checkpointing.bench_transports --transport pg \
        --size-gb 12 --device cuda
"""

from __future__ import annotations

import argparse
import time
from concurrent.futures import ThreadPoolExecutor
from datetime import timedelta
from typing import Dict

import torch
from torch.distributed import TCPStore


def make_state_dict(size_gb: float, device: torch.device, chunk_mb: int = 64) -> Dict[str, torch.Tensor]:
    n_chunks = max(1, int(size_gb * 1024 // chunk_mb))
    chunk_elems = chunk_mb * 1024 * 1024 // 2  # bf16
    return {
        f"w{i}": torch.randn(chunk_elems, dtype=torch.bfloat16, device=device)
        for i in range(n_chunks)
    }


def bench_pg(size_gb: float, device: torch.device) -> None:
    from torchft_amd.checkpointing.pg_transport import PGTransport
    from torchft_amd.process_group import ProcessGroupGloo, ProcessGroupRCCL

    store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
    addr = f"127.0.0.1:{store.port}/bench"
    sd = make_state_dict(size_gb, device)
    nbytes = sum(t.numel() * t.element_size() for t in sd.values())

    def run(rank: int) -> float:
        pg = (
            ProcessGroupRCCL(timeout=timedelta(seconds=300))
            if device.type == "cuda"
            else ProcessGroupGloo(timeout=timedelta(seconds=300))
        )
        pg.configure(addr, f"r{rank}", rank, 2)
        t = PGTransport(pg, timeout=timedelta(seconds=300), device=device)
        start = time.perf_counter()
        if rank == 0:
            t.send_checkpoint([1], step=1, state_dict=sd, timeout=timedelta(seconds=300))
        else:
            t.recv_checkpoint(src_rank=0, metadata="", step=1, timeout=timedelta(seconds=300))
        if device.type == "cuda":
            torch.cuda.synchronize()
        return time.perf_counter() - start

    with ThreadPoolExecutor(max_workers=2) as ex:
        times = list(ex.map(run, range(2)))
    elapsed = max(times)
    print(
        f"pg_transport: {nbytes / 1e9:.2f} GB in {elapsed:.2f}s = "
        f"{nbytes / elapsed / 1e9:.2f} GB/s"
    )


def bench_http(size_gb: float, device: torch.device, num_chunks: int = 0) -> None:
    from torchft_amd.checkpointing.http_transport import HTTPTransport

    sd = make_state_dict(size_gb, device)
    nbytes = sum(t.numel() * t.element_size() for t in sd.values())
    src = HTTPTransport(timeout=timedelta(seconds=300), num_chunks=num_chunks)
    dst = HTTPTransport(timeout=timedelta(seconds=300), num_chunks=num_chunks)
    try:
        t0 = time.perf_counter()
        src.send_checkpoint([1], step=1, state_dict=sd, timeout=timedelta(seconds=300))
        staged = time.perf_counter() - t0
        t0 = time.perf_counter()
        dst.recv_checkpoint(src_rank=0, metadata=src.metadata(), step=1,
                            timeout=timedelta(seconds=300))
        fetched = time.perf_counter() - t0
        print(
            f"http_transport: {nbytes / 1e9:.2f} GB staged in {staged:.2f}s "
            f"({nbytes / staged / 1e9:.2f} GB/s), fetched in {fetched:.2f}s "
            f"({nbytes / fetched / 1e9:.2f} GB/s)"
        )
    finally:
        src.shutdown()
        dst.shutdown()


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--transport", choices=["pg", "http"], default="pg")
    p.add_argument("--size-gb", type=float, default=12.0)
    p.add_argument("--device", default="cpu")
    p.add_argument("--num-chunks", type=int, default=0)
    args = p.parse_args()
    device = torch.device(args.device)
    if args.transport == "pg":
        bench_pg(args.size_gb, device)
    else:
        bench_http(args.size_gb, device, args.num_chunks)


if __name__ == "__main__":
    main()
