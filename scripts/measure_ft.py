"""Measure the FT-specific latencies on MI355X:

1. RCCL communicator reconfigure time (abort + re-init against a fresh
   store prefix) — the cost of a membership change.
2. Heal time: live checkpoint send+recv of a Llama-8B-sized state dict
   through both transports.
3. Per-step control-plane overhead: quorum + should_commit round trip.

Run on a GPU box:  python scripts/measure_ft.py [--model llama3_8b]
Writes a JSON summary to stdout.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time
from datetime import timedelta

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def measure_reconfigure(n: int = 5) -> dict:
    from torch.distributed import TCPStore

    from torchft_amd.process_group import ProcessGroupGloo, ProcessGroupRCCL

    store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
    use_cuda = torch.cuda.is_available()
    pg = ProcessGroupRCCL(timeout=timedelta(seconds=60)) if use_cuda else ProcessGroupGloo()
    times = []
    for i in range(n + 1):
        # membership changes are seconds apart in practice; give the
        # background reaper time to retire the previous communicator, or
        # this tight loop measures the old comm's ~500 ms abort (the join
        # at the top of configure) instead of the rebuild
        if pg._reaper is not None:
            pg._reaper.join()
        t0 = time.perf_counter()
        pg.configure(f"127.0.0.1:{store.port}/reconf_{i}", "r0", 0, 1)
        # first collective proves the communicator is live
        t = torch.ones(1024, device="cuda" if use_cuda else "cpu")
        from torch.distributed.distributed_c10d import AllreduceOptions, ReduceOp

        opts = AllreduceOptions()
        opts.reduceOp = ReduceOp.SUM
        pg.allreduce([t], opts).wait()
        if use_cuda:
            torch.cuda.synchronize()
        times.append(time.perf_counter() - t0)
    pg.shutdown()
    times = times[1:]  # first includes one-time init
    return {
        "reconfigure_ms_avg": sum(times) / len(times) * 1000,
        "reconfigure_ms_max": max(times) * 1000,
        "n": n,
    }


def _heal_fetch_child(metadata: str, conn) -> None:
    # separate process: real heals cross process boundaries (no shared GIL)
    import time as _t

    from torchft_amd.checkpointing.http_transport import HTTPTransport

    dst = HTTPTransport(timeout=timedelta(seconds=600))
    t0 = _t.perf_counter()
    dst.recv_checkpoint(src_rank=0, metadata=metadata, step=1,
                        timeout=timedelta(seconds=600))
    conn.send(_t.perf_counter() - t0)
    dst.shutdown()


def measure_heal(size_gb: float, device: torch.device) -> dict:
    import multiprocessing as mp

    from torchft_amd.checkpointing.bench_transports import make_state_dict
    from torchft_amd.checkpointing.http_transport import HTTPTransport

    sd = make_state_dict(size_gb, device)
    nbytes = sum(t.numel() * t.element_size() for t in sd.values())
    src = HTTPTransport(timeout=timedelta(seconds=600))
    try:
        t0 = time.perf_counter()
        src.send_checkpoint([1], step=1, state_dict=sd, timeout=timedelta(seconds=600))
        stage_s = time.perf_counter() - t0

        ctx = mp.get_context("spawn")
        parent, child = ctx.Pipe()
        proc = ctx.Process(target=_heal_fetch_child, args=(src.metadata(), child))
        proc.start()
        fetch_s = parent.recv()
        proc.join(timeout=30)
        return {
            "heal_bytes_gb": nbytes / 1e9,
            "stage_s": stage_s,
            "fetch_s": fetch_s,
            "heal_total_s": stage_s + fetch_s,
            "effective_gbps": nbytes / (stage_s + fetch_s) / 1e9,
        }
    finally:
        src.shutdown()


def measure_control_plane(n: int = 50) -> dict:
    import threading

    from torchft_amd._ftcore import LighthouseServer, ManagerClient, ManagerServer

    lh = LighthouseServer(bind="127.0.0.1:0", min_replicas=1, join_timeout_ms=10)
    mgr = ManagerServer(
        replica_id="perf0",
        lighthouse_addr=lh.address(),
        hostname="127.0.0.1",
        bind="127.0.0.1:0",
        store_addr="127.0.0.1:1",
        world_size=1,
        heartbeat_interval=timedelta(milliseconds=100),
        connect_timeout=timedelta(seconds=5),
    )
    try:
        c = ManagerClient(mgr.address(), connect_timeout=timedelta(seconds=5))
        qt, ct = [], []
        for i in range(n):
            t0 = time.perf_counter()
            c._quorum(
                group_rank=0, step=i, checkpoint_metadata="", shrink_only=False,
                timeout=timedelta(seconds=5),
            )
            qt.append(time.perf_counter() - t0)
            t0 = time.perf_counter()
            c.should_commit(0, i, True, timedelta(seconds=5))
            ct.append(time.perf_counter() - t0)
        qt.sort()
        ct.sort()
        return {
            "quorum_ms_p50": qt[n // 2] * 1000,
            "quorum_ms_p95": qt[int(n * 0.95)] * 1000,
            "should_commit_ms_p50": ct[n // 2] * 1000,
            "should_commit_ms_p95": ct[int(n * 0.95)] * 1000,
        }
    finally:
        mgr.shutdown()
        lh.shutdown()


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--heal-gb", type=float, default=16.0,
                   help="state-dict size (Llama-3-8B bf16 = 16 GB)")
    args = p.parse_args()
    device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")

    out = {"device": str(device)}
    out.update(measure_control_plane())
    out.update(measure_reconfigure())
    out.update(measure_heal(args.heal_gb if device.type == "cuda" else 0.5, device))
    print(json.dumps(out, indent=1))


if __name__ == "__main__":
    main()
