"""Break down RCCL communicator reconfigure latency on MI355X (world 1).

Round-1 measured abort + re-init + first collective at 546 ms total; this
script attributes it: abort, TCPStore client dial, PrefixStore wrap,
torch ProcessGroup ctor, RCCL backend ctor (nonblocking), eager connect,
first allreduce. Informs the in-place-reconfigure work (VERDICT item 9;
reference model: torchft/torchcomms.py:149-184 handle exchange).

Run on a GPU box: python scripts/measure_reconfigure.py
"""

import os
import sys
import time
from datetime import timedelta

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from torch.distributed import PrefixStore, ProcessGroup as BasePG, TCPStore
from torch.distributed.distributed_c10d import AllreduceOptions, ReduceOp

os.environ.setdefault("TORCH_NCCL_NONBLOCKING_TIMEOUT", "60")


def main() -> None:
    assert torch.cuda.is_available()
    torch.cuda.set_device(0)
    server = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)

    from torch.distributed import ProcessGroupNCCL as RCCL

    t_tensor = torch.ones(1024, device="cuda")

    def one_cycle(epoch: int, reuse_client: TCPStore | None):
        times = {}

        def mark(name, t0):
            times[name] = (time.perf_counter() - t0) * 1000
            return time.perf_counter()

        t0 = time.perf_counter()
        if reuse_client is None:
            client = TCPStore(
                "127.0.0.1", server.port, is_master=False,
                wait_for_workers=False, timeout=timedelta(seconds=30),
            )
        else:
            client = reuse_client
        t0 = mark("store_dial", t0)
        store = PrefixStore(f"q{epoch}", client)
        t0 = mark("prefix", t0)

        opts = RCCL.Options()
        opts.config.blocking = False
        pg = BasePG(store, 0, 1)
        pg._set_default_backend(BasePG.BackendType.NCCL)
        t0 = mark("pg_ctor", t0)
        backend = RCCL(store, 0, 1, opts)
        backend._set_sequence_number_for_group()
        t0 = mark("backend_ctor", t0)
        backend.eager_connect_single_device(torch.device("cuda", 0))
        t0 = mark("eager_connect", t0)
        pg._register_backend(torch.device("cuda"), BasePG.BackendType.NCCL, backend)
        ar = AllreduceOptions()
        ar.reduceOp = ReduceOp.SUM
        pg.allreduce([t_tensor], ar).wait()
        torch.cuda.synchronize()
        t0 = mark("first_allreduce", t0)

        t0 = time.perf_counter()
        backend.abort()
        mark("abort", t0)
        return times, client

    # warm once (lazy module init etc.)
    one_cycle(0, None)

    print("== fresh TCPStore dial per reconfigure ==")
    for e in range(1, 4):
        times, _ = one_cycle(e * 10, None)
        print({k: round(v, 1) for k, v in times.items()},
              "total_ms=", round(sum(times.values()), 1))

    print("== async abort (background thread) + cached client ==")
    import threading

    cached0 = TCPStore("127.0.0.1", server.port, is_master=False,
                       wait_for_workers=False, timeout=timedelta(seconds=30))
    prev_backend = None
    for e in range(1, 4):
        t0 = time.perf_counter()
        if prev_backend is not None:
            th = threading.Thread(target=prev_backend.abort)
            th.start()
        else:
            th = None
        store = PrefixStore(f"aq{e}", cached0)
        opts = RCCL.Options()
        opts.config.blocking = False
        pg = BasePG(store, 0, 1)
        pg._set_default_backend(BasePG.BackendType.NCCL)
        backend = RCCL(store, 0, 1, opts)
        backend._set_sequence_number_for_group()
        backend.eager_connect_single_device(torch.device("cuda", 0))
        pg._register_backend(torch.device("cuda"), BasePG.BackendType.NCCL, backend)
        ar = AllreduceOptions()
        ar.reduceOp = ReduceOp.SUM
        pg.allreduce([t_tensor], ar).wait()
        torch.cuda.synchronize()
        ready_ms = (time.perf_counter() - t0) * 1000
        if th is not None:
            th.join()
        total_ms = (time.perf_counter() - t0) * 1000
        print(f"new comm ready in {ready_ms:.1f} ms (old abort joined at "
              f"{total_ms:.1f} ms, off the critical path)")
        prev_backend = backend
    prev_backend.abort()

    print("== cached TCPStore client (new prefix only) ==")
    cached = TCPStore("127.0.0.1", server.port, is_master=False,
                      wait_for_workers=False, timeout=timedelta(seconds=30))
    for e in range(1, 4):
        times, _ = one_cycle(100 + e * 10, cached)
        print({k: round(v, 1) for k, v in times.items()},
              "total_ms=", round(sum(times.values()), 1))


if __name__ == "__main__":
    main()
