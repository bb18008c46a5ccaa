"""Cross-process should_commit barrier latency at world_size 2.

Round-1's BASELINE quoted 0.012 ms p50 for should_commit — a world-1
same-process loopback number. This measures the real thing: two rank
processes of one replica group, the ManagerServer aggregating over TCP,
p50/p95 over many barriers. Runs anywhere (CPU).
"""

from __future__ import annotations

import os
import statistics
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

RANK_CODE = r"""
import os, sys, time, statistics
sys.path.insert(0, os.environ["TFT_REPO"])
from datetime import timedelta
from torchft_amd._ftcore import ManagerClient, ManagerServer
from torch.distributed import TCPStore

rank = int(os.environ["TFT_RANK"])
world = 2
store = TCPStore("127.0.0.1", int(os.environ["TFT_STORE_PORT"]),
                 is_master=False, wait_for_workers=False)
if rank == 0:
    server = ManagerServer(
        replica_id="lat_measure",
        lighthouse_addr=os.environ["TORCHFT_LIGHTHOUSE"],
        hostname="127.0.0.1",
        bind="0.0.0.0:0",
        store_addr=f"127.0.0.1:{os.environ['TFT_STORE_PORT']}",
        world_size=world,
        heartbeat_interval=timedelta(milliseconds=100),
        connect_timeout=timedelta(seconds=10),
        quorum_retries=0,
    )
    store.set("mgr_addr", server.address())
addr = store.get("mgr_addr").decode()
client = ManagerClient(addr, connect_timeout=timedelta(seconds=10))

N = int(os.environ.get("TFT_ITERS", "300"))
lat = []
for step in range(N):
    t0 = time.perf_counter()
    ok = client.should_commit(rank, step, True, timeout=timedelta(seconds=10))
    lat.append((time.perf_counter() - t0) * 1000)
    assert ok
lat_sorted = sorted(lat[20:])  # drop warmup
p50 = statistics.median(lat_sorted)
p95 = lat_sorted[int(len(lat_sorted) * 0.95)]
print(f"RESULT rank={rank} p50={p50:.3f}ms p95={p95:.3f}ms", flush=True)
if rank == 0:
    time.sleep(1.0)
    server.shutdown()
"""


def main() -> None:
    from torch.distributed import TCPStore

    from torchft_amd._ftcore import LighthouseServer

    lighthouse = LighthouseServer(
        bind="127.0.0.1:0", min_replicas=1, join_timeout_ms=200
    )
    store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)

    env = dict(os.environ)
    env.update(
        {
            "TFT_REPO": REPO,
            "TFT_STORE_PORT": str(store.port),
            "TORCHFT_LIGHTHOUSE": lighthouse.address(),
        }
    )
    procs = []
    for rank in range(2):
        e = dict(env)
        e["TFT_RANK"] = str(rank)
        procs.append(
            subprocess.Popen([sys.executable, "-c", RANK_CODE], env=e)
        )
    deadline = time.time() + 120
    for p in procs:
        p.wait(timeout=max(1, deadline - time.time()))
    lighthouse.shutdown()
    sys.exit(max(p.returncode for p in procs))


if __name__ == "__main__":
    main()
