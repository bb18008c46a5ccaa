"""Job launcher: N fault-tolerant replica groups on one host (or a cluster
template), each its own torchrun, plus the lighthouse.

Reference parity: torchft/torchx.py (the ``hsdp`` torchX component spawning
replica-group torchrun roles with REPLICA_GROUP_ID / NUM_REPLICA_GROUPS
env) and torchft/examples/slurm. torchX isn't part of this image, so the
same component logic ships as a standalone CLI:

    python -m torchft_amd.launcher --replicas 2 --nproc-per-node 4 \
        --min-replicas 1 -- train_ddp.py --steps 100
"""

from __future__ import annotations

import argparse
import os
import signal
import subprocess
import sys
import time
from typing import List, Optional


def build_replica_cmd(
    script_and_args: List[str],
    replica_group_id: int,
    num_replica_groups: int,
    nproc_per_node: int,
    lighthouse_addr: str,
    master_port: int,
) -> tuple[List[str], dict]:
    """The command + env for one replica group's torchrun (the torchX
    component's role, as a plain subprocess)."""
    cmd = [
        sys.executable,
        "-m",
        "torch.distributed.run",
        "--nnodes=1",
        f"--nproc-per-node={nproc_per_node}",
        "--master-addr=127.0.0.1",
        f"--master-port={master_port}",
        *script_and_args,
    ]
    env = dict(os.environ)
    env.update(
        {
            "REPLICA_GROUP_ID": str(replica_group_id),
            "NUM_REPLICA_GROUPS": str(num_replica_groups),
            "TORCHFT_LIGHTHOUSE": lighthouse_addr,
        }
    )
    return cmd, env


def main(argv: Optional[List[str]] = None) -> int:
    parser = argparse.ArgumentParser(description=__doc__)
    parser.add_argument("--replicas", type=int, default=2)
    parser.add_argument("--nproc-per-node", type=int, default=1)
    parser.add_argument("--min-replicas", type=int, default=1)
    parser.add_argument("--base-port", type=int, default=29600)
    parser.add_argument("--join-timeout-ms", type=int, default=60000)
    parser.add_argument("script", nargs=argparse.REMAINDER,
                        help="-- script.py [script args]")
    args = parser.parse_args(argv)

    script = [s for s in args.script if s != "--"]
    if not script:
        parser.error("missing training script")

    from torchft_amd._ftcore import LighthouseServer

    lighthouse = LighthouseServer(
        bind="0.0.0.0:0",
        min_replicas=args.min_replicas,
        join_timeout_ms=args.join_timeout_ms,
    )
    print(f"[launcher] lighthouse at {lighthouse.address()}", flush=True)

    procs: List[subprocess.Popen] = []
    try:
        for g in range(args.replicas):
            cmd, env = build_replica_cmd(
                script, g, args.replicas, args.nproc_per_node,
                lighthouse.address(), args.base_port + g * 10,
            )
            print(f"[launcher] starting replica group {g}: {' '.join(cmd)}", flush=True)
            procs.append(subprocess.Popen(cmd, env=env))

        rc = 0
        while procs:
            time.sleep(1)
            for p in list(procs):
                code = p.poll()
                if code is not None:
                    procs.remove(p)
                    if code != 0:
                        rc = code
        return rc
    except KeyboardInterrupt:
        return 130
    finally:
        for p in procs:
            try:
                p.send_signal(signal.SIGTERM)
            except OSError:
                pass
        for p in procs:
            try:
                p.wait(timeout=10)
            except subprocess.TimeoutExpired:
                p.kill()
        lighthouse.shutdown()


if __name__ == "__main__":
    sys.exit(main())
