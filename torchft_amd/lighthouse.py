"""Lighthouse CLI (reference parity: the ``torchft_lighthouse`` binary,
src/bin/lighthouse.rs + LighthouseOpt flags).

    python -m torchft_amd.lighthouse --bind 0.0.0.0:29510 --min_replicas 2 \
        [--join_timeout_ms 60000] [--quorum_tick_ms 100] [--heartbeat_timeout_ms 5000]

Serves the quorum protocol plus the HTML dashboard (GET /status) and the
kill endpoint (POST /replica/:id/kill) on the same port.
"""

from __future__ import annotations

import argparse
import signal
import sys
import threading


def lighthouse_main(argv: list[str] | None = None) -> None:
    parser = argparse.ArgumentParser(description=__doc__)
    parser.add_argument("--bind", default="0.0.0.0:29510",
                        help="address to bind the server to")
    parser.add_argument("--min_replicas", type=int, required=True,
                        help="minimum number of replicas to form a quorum")
    parser.add_argument("--join_timeout_ms", type=int, default=60000,
                        help="how long to wait for heartbeating stragglers")
    parser.add_argument("--quorum_tick_ms", type=int, default=100,
                        help="quorum check interval while waiting")
    parser.add_argument("--heartbeat_timeout_ms", type=int, default=5000,
                        help="replica considered dead after this silence")
    args = parser.parse_args(argv)

    from torchft_amd._ftcore import LighthouseServer

    server = LighthouseServer(
        bind=args.bind,
        min_replicas=args.min_replicas,
        join_timeout_ms=args.join_timeout_ms,
        quorum_tick_ms=args.quorum_tick_ms,
        heartbeat_timeout_ms=args.heartbeat_timeout_ms,
    )
    print(f"lighthouse listening on {server.address()}", flush=True)

    stop = threading.Event()
    try:
        signal.signal(signal.SIGINT, lambda *a: stop.set())
        signal.signal(signal.SIGTERM, lambda *a: stop.set())
    except ValueError:
        pass  # not the main thread (embedded use) — rely on stop/exception
    stop.wait()
    server.shutdown()


if __name__ == "__main__":
    lighthouse_main(sys.argv[1:])
