"""Chaos harness: kill replicas through the lighthouse dashboard endpoints.

Reference parity: torchft/examples/slurm/punisher.py — kill_one / kill_all /
kill_loop with a mean-time-between-failures; the kill path goes
POST /replica/{id}/kill on the lighthouse, which forwards a kill RPC to the
replica's manager (which exits(1), exercising the full detection →
requorum → heal pipeline).

    python examples/chaos/punisher.py --lighthouse http://host:port kill-one
    python examples/chaos/punisher.py --lighthouse ... kill-loop --mtbf-secs 60
"""

from __future__ import annotations

import argparse
import random
import re
import sys
import time
import urllib.request


def list_replicas(lighthouse: str) -> list[str]:
    with urllib.request.urlopen(f"{lighthouse}/status", timeout=10) as resp:
        body = resp.read().decode()
    # replica ids appear in the kill-button form actions
    return re.findall(r'action="/replica/([^"]+)/kill"', body)


def kill(lighthouse: str, replica_id: str) -> None:
    req = urllib.request.Request(
        f"{lighthouse}/replica/{urllib.request.quote(replica_id)}/kill", data=b"",
        method="POST",
    )
    with urllib.request.urlopen(req, timeout=30) as resp:
        print(f"killed {replica_id}: {resp.status}", flush=True)


def kill_one(lighthouse: str) -> None:
    replicas = list_replicas(lighthouse)
    if not replicas:
        print("no replicas in quorum", file=sys.stderr)
        return
    kill(lighthouse, random.choice(replicas))


def kill_all(lighthouse: str) -> None:
    for rid in list_replicas(lighthouse):
        try:
            kill(lighthouse, rid)
        except Exception as e:  # noqa: BLE001
            print(f"kill {rid} failed: {e}", file=sys.stderr)


def kill_loop(lighthouse: str, mtbf_secs: float) -> None:
    """Poisson-ish kill loop: exponential inter-arrival with the given mean."""
    while True:
        delay = random.expovariate(1.0 / mtbf_secs)
        print(f"next kill in {delay:.1f}s", flush=True)
        time.sleep(delay)
        try:
            kill_one(lighthouse)
        except Exception as e:  # noqa: BLE001
            print(f"kill failed: {e}", file=sys.stderr)


def main() -> None:
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--lighthouse", required=True)
    sub = p.add_subparsers(dest="cmd", required=True)
    sub.add_parser("kill-one")
    sub.add_parser("kill-all")
    loop = sub.add_parser("kill-loop")
    loop.add_argument("--mtbf-secs", type=float, default=60.0)
    args = p.parse_args()

    if args.cmd == "kill-one":
        kill_one(args.lighthouse)
    elif args.cmd == "kill-all":
        kill_all(args.lighthouse)
    else:
        kill_loop(args.lighthouse, args.mtbf_secs)


if __name__ == "__main__":
    main()
