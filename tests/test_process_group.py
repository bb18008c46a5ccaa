"""Process-group layer tests (reference strategy: torchft/process_group_test.py
— per-backend collective table, threads-as-ranks with a port-0 TCPStore)."""

import threading
from concurrent.futures import ThreadPoolExecutor
from datetime import timedelta

import pytest
import torch
from torch.distributed import ReduceOp, TCPStore
from torch.distributed.distributed_c10d import (
    AllgatherOptions,
    AllreduceOptions,
    AllToAllOptions,
    BroadcastOptions,
    ReduceScatterOptions,
)

from torchft_amd.process_group import (
    ProcessGroupWrapper,
    ErrorSwallowingProcessGroupWrapper,
    FakeProcessGroupWrapper,
    ProcessGroupDummy,
    ProcessGroupGloo,
    _DummyWork,
)


class TestProcessGroupDummy:
    def test_allreduce_passthrough(self):
        pg = ProcessGroupDummy(0, 1)
        t = torch.ones(4)
        work = pg.allreduce([t], AllreduceOptions())
        assert work.wait()
        torch.testing.assert_close(t, torch.ones(4))

    def test_allgather_copies(self):
        pg = ProcessGroupDummy(0, 1)
        inp = [torch.arange(4.0)]
        out = [[torch.zeros(4)]]
        pg.allgather(out, inp, AllgatherOptions()).wait()
        torch.testing.assert_close(out[0][0], inp[0])

    def test_alltoall_copies(self):
        pg = ProcessGroupDummy(0, 1)
        inp = torch.arange(6.0)
        out = torch.zeros(6)
        pg.alltoall_base(out, inp, [], [], AllToAllOptions()).wait()
        torch.testing.assert_close(out, inp)

    def test_reduce_scatter_copies(self):
        pg = ProcessGroupDummy(0, 1)
        out = [torch.zeros(3)]
        inp = [[torch.arange(3.0)]]
        pg.reduce_scatter(out, inp, ReduceScatterOptions()).wait()
        torch.testing.assert_close(out[0], inp[0][0])

    def test_configure_counts(self):
        pg = ProcessGroupDummy(0, 1)
        pg.configure("addr", "0", 0, 1)
        assert pg.configure_count == 1

    def test_size_and_backend(self):
        pg = ProcessGroupDummy(0, 1)
        assert pg.size() == 1
        assert pg.getBackendName() == "torchft-dummy"


def _run_ranks(world_size, fn):
    """Threads-as-ranks helper with a shared port-0 TCPStore."""
    store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
    store_addr = f"127.0.0.1:{store.port}/test"
    with ThreadPoolExecutor(max_workers=world_size) as ex:
        futs = [ex.submit(fn, store_addr, rank, world_size) for rank in range(world_size)]
        return [f.result(timeout=60) for f in futs]


class TestProcessGroupGloo:
    def test_allreduce_two_ranks(self):
        def run(store_addr, rank, world):
            pg = ProcessGroupGloo(timeout=timedelta(seconds=20))
            pg.configure(store_addr, f"r{rank}", rank, world)
            t = torch.full((8,), float(rank + 1))
            opts = AllreduceOptions()
            opts.reduceOp = ReduceOp.SUM
            pg.allreduce([t], opts).wait()
            return t

        results = _run_ranks(2, run)
        for t in results:
            torch.testing.assert_close(t, torch.full((8,), 3.0))

    def test_broadcast_and_allgather(self):
        def run(store_addr, rank, world):
            pg = ProcessGroupGloo(timeout=timedelta(seconds=20))
            pg.configure(store_addr, f"r{rank}", rank, world)
            t = torch.full((4,), float(rank))
            opts = BroadcastOptions()
            opts.rootRank = 1
            pg.broadcast([t], opts).wait()

            out = [[torch.zeros(4) for _ in range(world)]]
            pg.allgather(out, [t], AllgatherOptions()).wait()
            return t, out

        for t, out in _run_ranks(2, run):
            torch.testing.assert_close(t, torch.full((4,), 1.0))
            for o in out[0]:
                torch.testing.assert_close(o, torch.full((4,), 1.0))

    def test_send_recv(self):
        def run(store_addr, rank, world):
            pg = ProcessGroupGloo(timeout=timedelta(seconds=20))
            pg.configure(store_addr, f"r{rank}", rank, world)
            if rank == 0:
                t = torch.arange(5.0)
                pg.send([t], 1, tag=7).wait()
                return t
            t = torch.zeros(5)
            pg.recv([t], 0, tag=7).wait()
            return t

        r = _run_ranks(2, run)
        torch.testing.assert_close(r[0], r[1])

    def test_reconfigure_shrinks_world(self):
        # reconfiguring with a new prefix builds a fresh communicator
        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)

        def run(rank, world):
            pg = ProcessGroupGloo(timeout=timedelta(seconds=20))
            pg.configure(f"127.0.0.1:{store.port}/q0", f"r{rank}", rank, world)
            t = torch.ones(2)
            opts = AllreduceOptions()
            opts.reduceOp = ReduceOp.SUM
            pg.allreduce([t], opts).wait()
            assert t[0].item() == world
            return pg

        with ThreadPoolExecutor(max_workers=2) as ex:
            pgs = list(ex.map(lambda r: run(r, 2), range(2)))

        # shrink to world of 1 (only rank 0 survives)
        pg0 = pgs[0]
        pg0.configure(f"127.0.0.1:{store.port}/q1", "r0", 0, 1)
        t = torch.ones(2)
        opts = AllreduceOptions()
        opts.reduceOp = ReduceOp.SUM
        pg0.allreduce([t], opts).wait()
        assert t[0].item() == 1

    def test_reduce_scatter_rejected(self):
        pg = ProcessGroupGloo()
        with pytest.raises(RuntimeError, match="does not support"):
            pg.reduce_scatter([], [], ReduceScatterOptions())


class TestWrappers:
    def test_error_swallowing(self):
        inner = ProcessGroupDummy(0, 1)
        pg = ErrorSwallowingProcessGroupWrapper(inner)
        t = torch.ones(2)
        work = pg.allreduce([t], AllreduceOptions())
        assert work.wait()
        assert pg.error() is None

        pg.report_error(RuntimeError("boom"))
        work = pg.allreduce([t], AllreduceOptions())
        assert isinstance(work, _DummyWork)
        # configure clears the error
        pg.configure("addr", "0", 0, 1)
        assert pg.error() is None

    def test_fake_future_error(self):
        inner = ProcessGroupDummy(0, 1)
        pg = FakeProcessGroupWrapper(inner)
        pg.report_future_error(RuntimeError("injected"))
        t = torch.ones(2)
        work = pg.allreduce([t], AllreduceOptions())
        fut = work.get_future()
        with pytest.raises(RuntimeError, match="injected"):
            fut.wait()


class TestAbortDiagnostics:
    """abort() must dump the flight-recorder op log."""

    def test_oplog_dump_on_abort(self, tmp_path, monkeypatch):
        import json
        import os

        from torch.distributed import TCPStore

        monkeypatch.setenv("TORCHFT_ABORT_DUMP_DIR", str(tmp_path))
        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
        pg = ProcessGroupGloo(timeout=timedelta(seconds=5))
        pg.configure(f"127.0.0.1:{store.port}/qd", "r0", 0, 1)
        t = torch.ones(8)
        pg.allreduce([t], ReduceOp.SUM).wait()
        pg.abort()  # errored abort dumps the log

        dumps = [p for p in os.listdir(tmp_path) if p.startswith("oplog_")]
        assert len(dumps) == 1
        records = json.loads((tmp_path / dumps[0]).read_text())
        assert any(r["op"] == "allreduce" for r in records)
        rec = [r for r in records if r["op"] == "allreduce"][-1]
        assert rec["tensors"] == [["float32", [8]]]
        assert rec["status"] == "in_flight"

    def test_no_dump_without_env(self, tmp_path, monkeypatch):
        monkeypatch.delenv("TORCHFT_ABORT_DUMP_DIR", raising=False)
        pg = ProcessGroupGloo(timeout=timedelta(seconds=5))
        assert pg._oplog.dump("x") is None


class TestBackgroundReaper:
    def test_configure_overlaps_old_abort(self):
        """configure() must not block on the old communicator's teardown
        (on MI355X the RCCL abort alone is ~507 ms) and must join the
        previous teardown before starting the next."""
        import time

        aborted = []

        class _SlowAbortPG:
            def __init__(self, tag):
                self.tag = tag

            def abort(self):
                time.sleep(0.3)
                aborted.append(self.tag)

        built = []

        class _PG(ProcessGroupWrapper):
            def _build(self, store, rank, world_size):
                built.append(len(built))
                return _SlowAbortPG(built[-1])

        from torch.distributed import TCPStore

        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
        pg = _PG(timeout=timedelta(seconds=5))
        pg.configure(f"127.0.0.1:{store.port}/ra", "r0", 0, 1)  # no old comm

        t0 = time.perf_counter()
        pg.configure(f"127.0.0.1:{store.port}/rb", "r0", 0, 1)
        elapsed = time.perf_counter() - t0
        assert elapsed < 0.25, f"configure blocked on the old abort ({elapsed:.2f}s)"
        assert pg._reaper is not None

        # third configure joins the in-flight teardown first
        pg.configure(f"127.0.0.1:{store.port}/rc", "r0", 0, 1)
        assert aborted == [0]  # first comm retired exactly once, in order
        pg.shutdown()
        assert aborted == [0, 1]


class TestOpLogRing:
    """The abort post-mortem ring buffer (flight-recorder analog)."""

    def test_wraparound_keeps_latest(self):
        from torchft_amd.process_group import _OpLog

        log = _OpLog(capacity=4)
        seqs = [log.record(f"op{i}", [torch.zeros(1)], quorum_id=1) for i in range(10)]
        snap = log.snapshot()
        assert len(snap) == 4
        assert [r["seq"] for r in snap] == seqs[-4:]
        assert [r["op"] for r in snap] == ["op6", "op7", "op8", "op9"]

    def test_mark_updates_status_even_after_wrap(self):
        from torchft_amd.process_group import _OpLog

        log = _OpLog(capacity=4)
        seqs = [log.record("allreduce", [torch.zeros(1)], 1) for _ in range(6)]
        log.mark(seqs[-1], "completed")
        log.mark(seqs[0], "completed")  # already evicted: silent no-op
        snap = log.snapshot()
        assert snap[-1]["status"] == "completed"
        assert all(r["status"] == "issued" for r in snap[:-1])

    def test_concurrent_record_is_sequential(self):
        import threading

        from torchft_amd.process_group import _OpLog

        log = _OpLog(capacity=256)
        out = []

        def worker():
            for _ in range(50):
                out.append(log.record("op", [], None))

        threads = [threading.Thread(target=worker) for _ in range(4)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        assert sorted(out) == list(range(1, 201))
        snap = log.snapshot()
        assert [r["seq"] for r in snap] == list(range(1, 201))

    def test_dump_unwritable_dir_is_nonfatal(self, monkeypatch):
        from torchft_amd.process_group import ABORT_DUMP_DIR_ENV, _OpLog

        monkeypatch.setenv(ABORT_DUMP_DIR_ENV, "/proc/definitely/not/writable")
        log = _OpLog()
        log.record("allreduce", [torch.zeros(2)], 1)
        assert log.dump("t") is None  # logged, not raised


class TestDeadlineWork:
    """_DeadlineWork guards the CPU-side wait: missing the deadline aborts
    the process group (the RCCL watchdog-replacement policy), in user space
    where the FT layer can observe it."""

    class _SlowWork(torch.distributed._Work):
        def __init__(self, block_s: float):
            super().__init__()
            self._block_s = block_s

        def wait(self, timeout=None):
            import time
            time.sleep(self._block_s)
            return True

        def get_future(self):
            fut = torch.futures.Future()
            fut.set_result(None)
            return fut

    class _AbortRecorder:
        def __init__(self):
            self.aborted = 0

        def abort(self):
            self.aborted += 1

    def test_expired_wait_aborts_pg(self):
        from datetime import timedelta

        from torchft_amd.process_group import _DeadlineWork

        pg = self._AbortRecorder()
        w = _DeadlineWork(pg, self._SlowWork(1.0), timedelta(milliseconds=100))
        assert w.wait()  # the slow wait returns, but the deadline fired
        import time
        time.sleep(0.3)  # timer callback runs on the executor thread
        assert pg.aborted >= 1

    def test_fast_wait_does_not_abort(self):
        from datetime import timedelta

        from torchft_amd.process_group import _DeadlineWork

        pg = self._AbortRecorder()
        w = _DeadlineWork(pg, self._SlowWork(0.0), timedelta(seconds=5))
        assert w.wait()
        import time
        time.sleep(0.2)
        assert pg.aborted == 0


class TestCollectiveSurfaceGloo:
    """Remaining collective-surface ops per backend (reference:
    process_group_test.py's per-op table)."""

    def test_broadcast_one(self):
        def run(store_addr, rank, world):
            pg = ProcessGroupGloo(timeout=timedelta(seconds=20))
            pg.configure(store_addr, f"r{rank}", rank, world)
            t = torch.full((4,), float(rank + 10))
            pg.broadcast_one(t, root=1).wait()
            return t

        for t in _run_ranks(2, run):
            torch.testing.assert_close(t, torch.full((4,), 11.0))

    def test_allreduce_coalesced(self):
        def run(store_addr, rank, world):
            pg = ProcessGroupGloo(timeout=timedelta(seconds=20))
            pg.configure(store_addr, f"r{rank}", rank, world)
            ts = [torch.full((3,), float(rank + 1)), torch.full((5,), 2.0 * (rank + 1))]
            from torch.distributed.distributed_c10d import AllreduceCoalescedOptions

            opts = AllreduceCoalescedOptions()
            opts.reduceOp = ReduceOp.SUM
            pg.allreduce_coalesced(ts, opts).wait()
            return ts

        for ts in _run_ranks(2, run):
            torch.testing.assert_close(ts[0], torch.full((3,), 3.0))
            torch.testing.assert_close(ts[1], torch.full((5,), 6.0))

    def test_barrier(self):
        from torch.distributed.distributed_c10d import BarrierOptions

        def run(store_addr, rank, world):
            pg = ProcessGroupGloo(timeout=timedelta(seconds=20))
            pg.configure(store_addr, f"r{rank}", rank, world)
            pg.barrier(BarrierOptions()).wait()
            return True

        assert all(_run_ranks(2, run))
