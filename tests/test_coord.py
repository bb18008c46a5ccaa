"""Unit tests for the C++ coordination core.

Mirrors the reference's Rust unit tests: quorum_compute pure-function cases
(src/lighthouse.rs:612-1297) and compute_quorum_results recovery assignment
(src/manager.rs:627-1219), plus e2e lighthouse+manager over real sockets.
"""

import threading
import time
from datetime import timedelta

import pytest

from torchft_amd import _ftcore as core

TD = timedelta


def member(rid, step=0, world_size=1, shrink_only=False, commit_failures=0):
    return core.QuorumMember(
        replica_id=rid,
        address=f"addr_{rid}",
        store_address=f"store_{rid}",
        step=step,
        world_size=world_size,
        shrink_only=shrink_only,
        commit_failures=commit_failures,
    )


class TestQuorumCompute:
    def test_min_replicas_not_met(self):
        res, reason = core.quorum_compute(
            participants_with_age=[(member("a"), 0)],
            heartbeat_ages={"a": 0},
            min_replicas=2,
        )
        assert res is None
        assert "min_replicas" in reason

    def test_basic_quorum(self):
        res, reason = core.quorum_compute(
            participants_with_age=[(member("a"), 0), (member("b"), 0)],
            heartbeat_ages={"a": 0, "b": 0},
            min_replicas=2,
            join_timeout_ms=0,
        )
        assert res is not None
        assert [m.replica_id for m in res] == ["a", "b"]

    def test_dead_heartbeat_filtered(self):
        res, reason = core.quorum_compute(
            participants_with_age=[(member("a"), 0), (member("b"), 0)],
            heartbeat_ages={"a": 0, "b": 10_000},  # b is stale
            min_replicas=2,
            heartbeat_timeout_ms=5000,
        )
        assert res is None

    def test_join_timeout_waits_for_stragglers(self):
        # c heartbeats but hasn't joined; within join timeout -> no quorum
        res, reason = core.quorum_compute(
            participants_with_age=[(member("a"), 100), (member("b"), 100)],
            heartbeat_ages={"a": 0, "b": 0, "c": 0},
            min_replicas=2,
            join_timeout_ms=60_000,
        )
        assert res is None
        assert "straggler" in reason
        # after the join timeout elapses -> quorum without c
        res, reason = core.quorum_compute(
            participants_with_age=[(member("a"), 70_000), (member("b"), 100)],
            heartbeat_ages={"a": 0, "b": 0, "c": 0},
            min_replicas=2,
            join_timeout_ms=60_000,
        )
        assert res is not None
        assert [m.replica_id for m in res] == ["a", "b"]

    def test_fast_quorum_skips_join_timeout(self):
        # previous members all healthy and joined -> immediate quorum even
        # though a straggler (c) is heartbeating
        res, reason = core.quorum_compute(
            participants_with_age=[(member("a"), 0), (member("b"), 0)],
            heartbeat_ages={"a": 0, "b": 0, "c": 0},
            prev_participants=[member("a"), member("b")],
            min_replicas=2,
            join_timeout_ms=60_000,
        )
        assert res is not None
        assert "Fast quorum" in reason

    def test_split_brain_guard(self):
        # 5 alive, only 2 participating: 2 <= 5//2 -> no quorum
        res, reason = core.quorum_compute(
            participants_with_age=[(member("a"), 100_000), (member("b"), 100_000)],
            heartbeat_ages={k: 0 for k in "abcde"},
            min_replicas=1,
            join_timeout_ms=1,
        )
        assert res is None
        assert "half" in reason

    def test_shrink_only_excludes_newcomers(self):
        res, reason = core.quorum_compute(
            participants_with_age=[
                (member("a", shrink_only=True), 100_000),
                (member("b"), 100_000),
                (member("c"), 100_000),  # newcomer, not in prev quorum
            ],
            heartbeat_ages={"a": 0, "b": 0, "c": 0},
            prev_participants=[member("a"), member("b"), member("z")],
            min_replicas=1,
            join_timeout_ms=1,
        )
        assert res is not None
        assert [m.replica_id for m in res] == ["a", "b"]


def quorum_of(*members):
    q = core.Quorum()
    q.quorum_id = 5
    q.participants = list(members)
    return q


class TestComputeQuorumResults:
    def test_all_healthy_same_step(self):
        q = quorum_of(member("a", step=3), member("b", step=3))
        ra = core.compute_quorum_results("a", 0, q, True)
        assert not ra.heal
        assert ra.replica_rank == 0
        assert ra.replica_world_size == 2
        assert ra.max_step == 3
        assert ra.max_world_size == 2
        assert ra.recover_dst_replica_ranks == []
        assert ra.replica_ids == ["a", "b"]

    def test_behind_replica_heals(self):
        q = quorum_of(member("a", step=5), member("b", step=3))
        rb = core.compute_quorum_results("b", 0, q, True)
        assert rb.heal
        assert rb.recover_src_replica_rank == 0
        assert rb.recover_src_manager_address == "addr_a"
        assert rb.max_step == 5
        assert rb.max_replica_rank is None  # b not at max step
        ra = core.compute_quorum_results("a", 0, q, True)
        assert not ra.heal
        assert ra.recover_dst_replica_ranks == [1]
        assert ra.max_replica_rank == 0

    def test_init_sync_forces_recovery_at_step0(self):
        q = quorum_of(member("a", step=0), member("b", step=0))
        # group_rank 0: primary is max_participants[0] == a -> b heals from a
        ra = core.compute_quorum_results("a", 0, q, True)
        assert not ra.heal
        assert ra.recover_dst_replica_ranks == [1]
        rb = core.compute_quorum_results("b", 0, q, True)
        assert rb.heal
        assert rb.recover_src_replica_rank == 0

    def test_no_init_sync_no_recovery_at_step0(self):
        q = quorum_of(member("a", step=0), member("b", step=0))
        for rid in ("a", "b"):
            r = core.compute_quorum_results(rid, 0, q, False)
            assert not r.heal
            assert r.recover_dst_replica_ranks == []

    def test_round_robin_assignment_offsets_by_group_rank(self):
        # 2 up-to-date (a, b at step 4), 2 behind (c, d at step 2)
        q = quorum_of(
            member("a", step=4),
            member("b", step=4),
            member("c", step=2),
            member("d", step=2),
        )
        # sorted: a(0) b(1) c(2) d(3); dst=[2,3], up_to_date=[0,1]
        # group_rank 0: c->a, d->b
        assert core.compute_quorum_results("c", 0, q, True).recover_src_replica_rank == 0
        assert core.compute_quorum_results("d", 0, q, True).recover_src_replica_rank == 1
        # group_rank 1 rotates: c->b, d->a
        assert core.compute_quorum_results("c", 1, q, True).recover_src_replica_rank == 1
        assert core.compute_quorum_results("d", 1, q, True).recover_src_replica_rank == 0
        # senders see their dst lists
        assert core.compute_quorum_results("a", 0, q, True).recover_dst_replica_ranks == [2]
        assert core.compute_quorum_results("b", 0, q, True).recover_dst_replica_ranks == [3]

    def test_store_address_rotates_with_group_rank(self):
        q = quorum_of(member("a", step=1), member("b", step=1))
        assert core.compute_quorum_results("a", 0, q, True).store_address == "store_a"
        assert core.compute_quorum_results("a", 1, q, True).store_address == "store_b"

    def test_commit_failures_propagate_max(self):
        q = quorum_of(member("a", commit_failures=2), member("b", commit_failures=0))
        assert core.compute_quorum_results("a", 0, q, False).commit_failures == 2

    def test_not_in_quorum_raises(self):
        q = quorum_of(member("a"))
        with pytest.raises(RuntimeError, match="not participating"):
            core.compute_quorum_results("zzz", 0, q, True)


class TestEndToEnd:
    def test_two_replica_quorum_and_commit(self):
        lh = core.LighthouseServer(bind="127.0.0.1:0", min_replicas=2, join_timeout_ms=100)
        mgrs = [
            core.ManagerServer(
                replica_id=f"rep{i}",
                lighthouse_addr=lh.address(),
                hostname="127.0.0.1",
                bind="127.0.0.1:0",
                store_addr=f"store{i}",
                world_size=1,
                heartbeat_interval=TD(milliseconds=50),
                connect_timeout=TD(seconds=5),
            )
            for i in range(2)
        ]
        try:
            results = {}

            def quorum(i):
                c = core.ManagerClient(mgrs[i].address(), connect_timeout=TD(seconds=5))
                results[i] = c._quorum(
                    group_rank=0,
                    step=0,
                    checkpoint_metadata=f"meta{i}",
                    shrink_only=False,
                    timeout=TD(seconds=10),
                )

            ts = [threading.Thread(target=quorum, args=(i,)) for i in range(2)]
            [t.start() for t in ts]
            [t.join(15) for t in ts]
            assert len(results) == 2
            assert results[0].quorum_id == results[1].quorum_id
            assert results[0].replica_ids == ["rep0", "rep1"]
            assert results[0].replica_rank == 0
            assert results[1].replica_rank == 1

            # healthy group commits; a reported failure vetoes the group
            c0 = core.ManagerClient(mgrs[0].address(), connect_timeout=TD(seconds=5))
            assert c0.should_commit(0, 1, True, TD(seconds=5)) is True
            assert c0.should_commit(0, 2, False, TD(seconds=5)) is False

            # checkpoint metadata lookup
            assert c0._checkpoint_metadata(0, TD(seconds=5)) == "meta0"
            with pytest.raises(Exception):
                c0._checkpoint_metadata(99, TD(seconds=5))
        finally:
            for m in mgrs:
                m.shutdown()
            lh.shutdown()

    def test_lighthouse_client_direct(self):
        lh = core.LighthouseServer(bind="127.0.0.1:0", min_replicas=1, join_timeout_ms=10)
        try:
            c = core.LighthouseClient(lh.address(), connect_timeout=TD(seconds=5))
            c.heartbeat("solo")
            q = c.quorum(
                replica_id="solo",
                timeout=TD(seconds=10),
                address="addr",
                store_address="store",
                step=7,
                world_size=2,
            )
            assert q.quorum_id >= 1
            assert len(q.participants) == 1
            assert q.participants[0].replica_id == "solo"
            assert q.participants[0].step == 7
        finally:
            lh.shutdown()

    def test_quorum_timeout_raises(self):
        lh = core.LighthouseServer(bind="127.0.0.1:0", min_replicas=2)
        try:
            c = core.LighthouseClient(lh.address(), connect_timeout=TD(seconds=5))
            with pytest.raises(TimeoutError):
                c.quorum(replica_id="lonely", timeout=TD(milliseconds=300))
        finally:
            lh.shutdown()

    def test_http_status_page(self):
        import urllib.request

        lh = core.LighthouseServer(bind="127.0.0.1:0", min_replicas=1, join_timeout_ms=10)
        try:
            addr = lh.address()
            with urllib.request.urlopen(addr.replace("http://", "http://") + "/status", timeout=5) as r:
                body = r.read().decode()
            assert "Lighthouse" in body
        finally:
            lh.shutdown()


class TestResilienceCoord:
    def test_lighthouse_restart_heartbeat_reconnects(self):
        """The manager's heartbeat/quorum clients reconnect with backoff
        when the lighthouse restarts (reference: create_lighthouse_client
        retry path)."""
        import time

        lh = core.LighthouseServer(bind="127.0.0.1:0", min_replicas=1, join_timeout_ms=10)
        port = int(lh.address().rsplit(":", 1)[1])
        mgr = core.ManagerServer(
            replica_id="restart0",
            lighthouse_addr=lh.address(),
            hostname="127.0.0.1",
            bind="127.0.0.1:0",
            store_addr="s",
            world_size=1,
            heartbeat_interval=TD(milliseconds=50),
            connect_timeout=TD(seconds=2),
        )
        try:
            c = core.ManagerClient(mgr.address(), connect_timeout=TD(seconds=5))
            r = c._quorum(0, 0, "", False, TD(seconds=5))
            assert r.quorum_id >= 1

            # kill the lighthouse, then bring a new one up on the SAME port
            lh.shutdown()
            time.sleep(0.3)
            lh = core.LighthouseServer(
                bind=f"127.0.0.1:{port}", min_replicas=1, join_timeout_ms=10
            )
            # the manager must re-reach the new lighthouse and form a quorum
            r = c._quorum(0, 1, "", False, TD(seconds=10))
            assert r.max_step >= 0
        finally:
            mgr.shutdown()
            lh.shutdown()

    def test_quorum_times_out_when_rank_missing(self):
        lh = core.LighthouseServer(bind="127.0.0.1:0", min_replicas=1, join_timeout_ms=10)
        mgr = core.ManagerServer(
            replica_id="missing0",
            lighthouse_addr=lh.address(),
            hostname="127.0.0.1",
            bind="127.0.0.1:0",
            store_addr="s",
            world_size=2,  # second rank never joins
            heartbeat_interval=TD(milliseconds=50),
            connect_timeout=TD(seconds=2),
        )
        try:
            c = core.ManagerClient(mgr.address(), connect_timeout=TD(seconds=5))
            with pytest.raises(TimeoutError):
                c._quorum(0, 0, "", False, TD(milliseconds=500))
        finally:
            mgr.shutdown()
            lh.shutdown()


class TestLighthouseFailover:
    def test_manager_survives_lighthouse_restart(self):
        """Kill the lighthouse and bring a new one up at the SAME address:
        the manager's retry/backoff client must reconnect and the next
        quorum succeed (managers are lighthouse-restart tolerant)."""
        import socket

        # reserve a port so the replacement can bind the same address
        probe = socket.socket()
        probe.bind(("127.0.0.1", 0))
        port = probe.getsockname()[1]
        probe.close()

        lh = core.LighthouseServer(
            bind=f"127.0.0.1:{port}", min_replicas=1, join_timeout_ms=100
        )
        mgr = core.ManagerServer(
            replica_id="fo0",
            lighthouse_addr=lh.address(),
            hostname="127.0.0.1",
            bind="127.0.0.1:0",
            store_addr="s",
            world_size=1,
            heartbeat_interval=TD(milliseconds=50),
            connect_timeout=TD(seconds=5),
            quorum_retries=3,
        )
        lh2 = None
        try:
            c = core.ManagerClient(mgr.address(), connect_timeout=TD(seconds=5))
            q1 = c._quorum(0, 0, "m", False, TD(seconds=10))
            assert q1.quorum_id >= 0

            lh.shutdown()
            time.sleep(0.3)
            lh2 = core.LighthouseServer(
                bind=f"127.0.0.1:{port}", min_replicas=1, join_timeout_ms=100
            )
            q2 = c._quorum(0, 1, "m", False, TD(seconds=15))
            assert q2.quorum_id >= 0
            assert q2.replica_ids == ["fo0"]
        finally:
            mgr.shutdown()
            if lh2 is not None:
                lh2.shutdown()


class TestWaiterSelfHeartbeat:
    def test_blocked_quorum_request_does_not_starve_itself(self):
        """A replica blocked in the quorum RPC longer than
        heartbeat_timeout_ms must not go stale itself: the pending request
        counts as liveness (the waiter refreshes its own heartbeat each
        wait slice). Regression: with 16 stale straggler heartbeats
        blocking the split-brain guard, the waiter starved and timed out
        even after the stragglers expired."""
        import time
        from concurrent.futures import ThreadPoolExecutor

        lh = core.LighthouseServer(
            bind="127.0.0.1:0", min_replicas=1, join_timeout_ms=10
        )
        try:
            def hammer(i):
                c = core.LighthouseClient(
                    lh.address(), connect_timeout=timedelta(seconds=10)
                )
                for _ in range(20):
                    c.heartbeat(f"ghost_{i}")

            with ThreadPoolExecutor(16) as ex:
                list(ex.map(hammer, range(16)))

            # request NOW, while the ghosts are still fresh: the guard
            # blocks (1 participant vs 17 heartbeating) until they expire
            # at heartbeat_timeout (5 s default) — then the waiter must
            # still be healthy enough to form the quorum
            c = core.LighthouseClient(
                lh.address(), connect_timeout=timedelta(seconds=10)
            )
            c.heartbeat("survivor")
            t0 = time.monotonic()
            q = c.quorum(replica_id="survivor", timeout=timedelta(seconds=12))
            elapsed = time.monotonic() - t0
            assert len(q.participants) == 1
            assert q.participants[0].replica_id == "survivor"
            assert 4.0 < elapsed < 8.0  # formed right after ghost expiry
        finally:
            lh.shutdown()


class TestCommitBarrierStress:
    def test_sixty_rounds_random_dissenter(self):
        """should_commit is an all-ranks vote: any dissenting rank vetoes
        the whole group, every rank sees the same verdict, and verdicts
        from consecutive rounds never bleed into each other."""
        import random

        lh = core.LighthouseServer(
            bind="127.0.0.1:0", min_replicas=1, join_timeout_ms=100
        )
        mgr = core.ManagerServer(
            replica_id="rep0", lighthouse_addr=lh.address(),
            hostname="127.0.0.1", bind="127.0.0.1:0", store_addr="store0",
            world_size=4, heartbeat_interval=TD(milliseconds=50),
            connect_timeout=TD(seconds=5),
        )
        try:
            rng = random.Random(0)
            for step in range(1, 61):
                dissenter = rng.randrange(5)  # 4 = unanimous round
                results = [None] * 4

                def vote(r):
                    c = core.ManagerClient(
                        mgr.address(), connect_timeout=TD(seconds=5)
                    )
                    results[r] = c.should_commit(
                        r, step, r != dissenter, TD(seconds=10)
                    )

                ts = [threading.Thread(target=vote, args=(r,)) for r in range(4)]
                [t.start() for t in ts]
                [t.join(15) for t in ts]
                expected = dissenter == 4
                assert all(v == expected for v in results), (
                    step, dissenter, results
                )
        finally:
            mgr.shutdown()
            lh.shutdown()


class TestCoordResourceHygiene:
    def test_server_lifecycle_no_fd_leak(self):
        import os

        def nfds():
            return len(os.listdir(f"/proc/{os.getpid()}/fd"))

        base = nfds()
        for _ in range(10):
            lh = core.LighthouseServer(
                bind="127.0.0.1:0", min_replicas=1, join_timeout_ms=10
            )
            c = core.LighthouseClient(lh.address(), connect_timeout=TD(seconds=5))
            c.heartbeat("x")
            q = c.quorum(replica_id="x", timeout=TD(seconds=5))
            assert len(q.participants) == 1
            lh.shutdown()
        assert nfds() <= base + 5, "fd leak across server lifecycles"

    def test_client_churn_no_fd_leak(self):
        import os

        def nfds():
            return len(os.listdir(f"/proc/{os.getpid()}/fd"))

        lh = core.LighthouseServer(
            bind="127.0.0.1:0", min_replicas=1, join_timeout_ms=10
        )
        try:
            base = nfds()
            for i in range(100):
                c = core.LighthouseClient(
                    lh.address(), connect_timeout=TD(seconds=5)
                )
                c.heartbeat(f"churn_{i % 4}")
                del c
            assert nfds() <= base + 5, "fd leak across client churn"
        finally:
            lh.shutdown()


class TestQuorumUserData:
    def test_member_data_round_trips(self):
        """QuorumMember.data (the JSON user dict, reference
        torchft.proto QuorumMember.data) must survive the quorum
        round-trip so custom FT algorithms can piggyback metadata."""
        import json

        lh = core.LighthouseServer(
            bind="127.0.0.1:0", min_replicas=2, join_timeout_ms=200
        )
        try:
            payloads = {
                "a": json.dumps({"zone": "az1", "slot": 3}),
                "b": json.dumps({"zone": "az2"}),
            }
            results = {}

            def join(rid):
                c = core.LighthouseClient(
                    lh.address(), connect_timeout=TD(seconds=5)
                )
                results[rid] = c.quorum(
                    replica_id=rid, timeout=TD(seconds=10), data=payloads[rid]
                )

            ts = [threading.Thread(target=join, args=(r,)) for r in ("a", "b")]
            [t.start() for t in ts]
            [t.join(15) for t in ts]
            assert set(results) == {"a", "b"}
            for q in results.values():
                by_id = {m.replica_id: m for m in q.participants}
                assert json.loads(by_id["a"].data) == {"zone": "az1", "slot": 3}
                assert json.loads(by_id["b"].data) == {"zone": "az2"}
        finally:
            lh.shutdown()


class TestDashboardKill:
    def test_kill_endpoint_terminates_manager(self):
        """POST /replica/{id}/kill on the lighthouse forwards a kill RPC to
        that replica's manager, which _exit(1)s (reference: the dashboard
        kill button -> Manager.kill -> process::exit)."""
        import subprocess
        import sys
        import urllib.request

        lh = core.LighthouseServer(
            bind="127.0.0.1:0", min_replicas=1, join_timeout_ms=100
        )
        p = None
        try:
            child_code = f'''
import time
from datetime import timedelta as TD
from torchft_amd import _ftcore as core
mgr = core.ManagerServer(replica_id="victim", lighthouse_addr="{lh.address()}",
    hostname="127.0.0.1", bind="127.0.0.1:0", store_addr="s", world_size=1,
    heartbeat_interval=TD(milliseconds=50), connect_timeout=TD(seconds=5))
c = core.ManagerClient(mgr.address(), connect_timeout=TD(seconds=5))
c._quorum(group_rank=0, step=0, checkpoint_metadata="m", shrink_only=False,
          timeout=TD(seconds=10))
print("in quorum", flush=True)
time.sleep(60)
'''
            p = subprocess.Popen(
                [sys.executable, "-c", child_code],
                stdout=subprocess.PIPE, text=True,
            )
            assert p.stdout.readline().strip() == "in quorum"
            req = urllib.request.Request(
                lh.address() + "/replica/victim/kill", data=b"", method="POST"
            )
            with urllib.request.urlopen(req, timeout=10) as r:
                assert r.status == 200
            assert p.wait(timeout=15) == 1  # manager _exit(1) on kill
            p = None
        finally:
            if p is not None:
                p.kill()
                p.wait(timeout=10)
            lh.shutdown()

    def test_kill_unknown_replica_is_404(self):
        import urllib.error
        import urllib.request

        lh = core.LighthouseServer(
            bind="127.0.0.1:0", min_replicas=1, join_timeout_ms=100
        )
        try:
            req = urllib.request.Request(
                lh.address() + "/replica/nobody/kill", data=b"", method="POST"
            )
            with pytest.raises(urllib.error.HTTPError):
                urllib.request.urlopen(req, timeout=10)
        finally:
            lh.shutdown()
