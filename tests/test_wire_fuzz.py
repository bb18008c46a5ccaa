"""Robustness of the C++ coordination servers against malformed input.

A public TCP port will see garbage (port scanners, half-open HTTP, truncated
frames); none of it may crash or wedge the lighthouse/manager.
"""

import random
import socket
import struct
from datetime import timedelta

from torchft_amd import _ftcore as core


def _port(addr: str) -> int:
    return int(addr.rsplit(":", 1)[1])


def _send_raw(port: int, payload: bytes) -> None:
    s = socket.create_connection(("127.0.0.1", port), timeout=5)
    try:
        s.sendall(payload)
        s.settimeout(0.5)
        try:
            s.recv(4096)
        except (socket.timeout, ConnectionResetError):
            pass  # server dropping a garbage connection is correct behavior
    finally:
        s.close()


class TestWireFuzz:
    def test_lighthouse_survives_garbage(self):
        lh = core.LighthouseServer(bind="127.0.0.1:0", min_replicas=1, join_timeout_ms=10)
        port = _port(lh.address())
        rng = random.Random(0)
        try:
            payloads = [
                b"",                                     # connect + close
                b"\x00" * 5,                             # zero-length frame
                struct.pack("<I", 0xFFFFFFFF) + b"\x01",  # oversized frame
                b"GET /nope HTTP/1.1\r\n\r\n",           # unknown http path
                b"POST /replica/ghost/kill HTTP/1.1\r\n\r\n",  # kill unknown
                struct.pack("<I", 10) + bytes([99]) + b"short",  # unknown type, truncated
                bytes(rng.randrange(256) for _ in range(64)),     # pure noise
            ]
            for p in payloads:
                _send_raw(port, p)
            # the server still works after all of that
            c = core.LighthouseClient(lh.address(), connect_timeout=timedelta(seconds=5))
            c.heartbeat("fuzz_survivor")
            q = c.quorum(replica_id="fuzz_survivor", timeout=timedelta(seconds=10))
            assert len(q.participants) == 1
        finally:
            lh.shutdown()

    def test_manager_survives_garbage(self):
        lh = core.LighthouseServer(bind="127.0.0.1:0", min_replicas=1, join_timeout_ms=10)
        mgr = core.ManagerServer(
            replica_id="fuzz0",
            lighthouse_addr=lh.address(),
            hostname="127.0.0.1",
            bind="127.0.0.1:0",
            store_addr="s",
            world_size=1,
            heartbeat_interval=timedelta(milliseconds=100),
            connect_timeout=timedelta(seconds=2),
        )
        port = _port(mgr.address())
        try:
            for p in [b"\x00" * 5, b"junkjunkjunk", struct.pack("<I", 6) + bytes([5]) + b"tiny!"]:
                _send_raw(port, p)
            c = core.ManagerClient(mgr.address(), connect_timeout=timedelta(seconds=5))
            r = c._quorum(0, 0, "", False, timedelta(seconds=10))
            assert r.replica_world_size == 1
        finally:
            mgr.shutdown()
            lh.shutdown()

    def test_lighthouse_serves_while_connections_stall(self):
        """Slowloris-style: stalled half-open connections must not block
        real clients (per-connection threads, no single accept loop wedge)."""
        lh = core.LighthouseServer(bind="127.0.0.1:0", min_replicas=1, join_timeout_ms=10)
        port = _port(lh.address())
        conns = []
        try:
            for _ in range(5):
                s = socket.create_connection(("127.0.0.1", port), timeout=5)
                s.sendall(b"\x10")  # first byte of a length prefix, then stall
                conns.append(s)
            c = core.LighthouseClient(lh.address(), connect_timeout=timedelta(seconds=5))
            c.heartbeat("live")
            q = c.quorum(replica_id="live", timeout=timedelta(seconds=10))
            assert len(q.participants) == 1
        finally:
            for s in conns:
                s.close()
            lh.shutdown()

    def test_lighthouse_survives_random_payload_storm(self):
        """Property-style fuzz: 30 random payloads (varied length, some with
        plausible length prefixes) then a liveness check."""
        lh = core.LighthouseServer(bind="127.0.0.1:0", min_replicas=1, join_timeout_ms=10)
        port = _port(lh.address())
        rng = random.Random(1234)
        try:
            for i in range(30):
                n = rng.randrange(0, 256)
                body = bytes(rng.randrange(256) for _ in range(n))
                if i % 3 == 0 and n >= 4:
                    # plausible length prefix pointing past the real payload
                    body = struct.pack("<I", rng.randrange(0, 1 << 20)) + body[4:]
                _send_raw(port, body)
            c = core.LighthouseClient(lh.address(), connect_timeout=timedelta(seconds=5))
            c.heartbeat("storm_survivor")
            q = c.quorum(replica_id="storm_survivor", timeout=timedelta(seconds=10))
            assert len(q.participants) == 1
        finally:
            lh.shutdown()
