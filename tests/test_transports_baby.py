"""PGTransport + Baby process group tests (CPU/gloo)."""

from concurrent.futures import ThreadPoolExecutor
from datetime import timedelta

import pytest
import torch
from torch.distributed import TCPStore
from torch.distributed.distributed_c10d import AllreduceOptions, ReduceOp

from torchft_amd.baby_process_group import ProcessGroupBabyGloo
from torchft_amd.checkpointing.pg_transport import PGTransport
from torchft_amd.process_group import ProcessGroupGloo


class TestPGTransport:
    def test_send_recv_roundtrip(self):
        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
        addr = f"127.0.0.1:{store.port}/pgt"

        sd = {
            "w": torch.randn(64, 32),
            "b": torch.arange(10, dtype=torch.int64),
            "nested": {"x": torch.randn(5, 5).to(torch.bfloat16)},
            "step": 7,
        }

        def run(rank):
            pg = ProcessGroupGloo(timeout=timedelta(seconds=20))
            pg.configure(addr, f"r{rank}", rank, 2)
            t = PGTransport(pg, timeout=timedelta(seconds=20), device=torch.device("cpu"))
            if rank == 0:
                t.send_checkpoint([1], step=3, state_dict=sd, timeout=timedelta(seconds=20))
                return sd
            return t.recv_checkpoint(
                src_rank=0, metadata="<n/a>", step=3, timeout=timedelta(seconds=20)
            )

        with ThreadPoolExecutor(max_workers=2) as ex:
            sent, got = list(ex.map(run, range(2)))
        torch.testing.assert_close(got["w"], sd["w"])
        torch.testing.assert_close(got["b"], sd["b"])
        torch.testing.assert_close(got["nested"]["x"], sd["nested"]["x"])
        assert got["step"] == 7

    def test_inplace_recv(self):
        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
        addr = f"127.0.0.1:{store.port}/pgt2"
        src_sd = {"w": torch.randn(16, 16)}
        dst_w = torch.zeros(16, 16)

        def run(rank):
            pg = ProcessGroupGloo(timeout=timedelta(seconds=20))
            pg.configure(addr, f"r{rank}", rank, 2)
            if rank == 0:
                t = PGTransport(pg, timeout=timedelta(seconds=20), device=torch.device("cpu"))
                t.send_checkpoint([1], step=1, state_dict=src_sd, timeout=timedelta(seconds=20))
                return None
            t = PGTransport(
                pg,
                timeout=timedelta(seconds=20),
                device=torch.device("cpu"),
                state_dict=lambda: {"w": dst_w},
            )
            return t.recv_checkpoint(
                src_rank=0, metadata="<n/a>", step=1, timeout=timedelta(seconds=20)
            )

        with ThreadPoolExecutor(max_workers=2) as ex:
            _, got = list(ex.map(run, range(2)))
        # received in place: the provided destination tensor holds the data
        torch.testing.assert_close(dst_w, src_sd["w"])
        assert got["w"] is dst_w


class TestBabyGloo:
    def test_allreduce_through_subprocess(self):
        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
        addr = f"127.0.0.1:{store.port}/baby"

        def run(rank):
            pg = ProcessGroupBabyGloo(timeout=60.0)
            pg.configure(addr, f"r{rank}", rank, 2)
            try:
                t = torch.full((8,), float(rank + 1))
                opts = AllreduceOptions()
                opts.reduceOp = ReduceOp.SUM
                work = pg.allreduce([t], opts)
                work.wait()
                return t.clone()
            finally:
                pg.shutdown()

        with ThreadPoolExecutor(max_workers=2) as ex:
            results = list(ex.map(run, range(2)))
        for t in results:
            torch.testing.assert_close(t, torch.full((8,), 3.0))

    def test_future_completion(self):
        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
        addr = f"127.0.0.1:{store.port}/baby2"

        def run(rank):
            pg = ProcessGroupBabyGloo(timeout=60.0)
            pg.configure(addr, f"r{rank}", rank, 2)
            try:
                t = torch.full((4,), float(rank))
                opts = AllreduceOptions()
                opts.reduceOp = ReduceOp.SUM
                work = pg.allreduce([t], opts)
                fut = work.get_future()
                fut.wait()
                return t.clone()
            finally:
                pg.shutdown()

        with ThreadPoolExecutor(max_workers=2) as ex:
            results = list(ex.map(run, range(2)))
        for t in results:
            torch.testing.assert_close(t, torch.full((4,), 1.0))

    def test_reconfigure_respawns(self):
        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)

        def run(rank, prefix, world):
            pg = ProcessGroupBabyGloo(timeout=60.0)
            pg.configure(f"127.0.0.1:{store.port}/{prefix}", f"r{rank}", rank, world)
            return pg

        with ThreadPoolExecutor(max_workers=2) as ex:
            pgs = list(ex.map(lambda r: run(r, "q0", 2), range(2)))
        try:
            # shrink to world 1 — kill-and-respawn with new store prefix
            pgs[0].configure(f"127.0.0.1:{store.port}/q1", "r0", 0, 1)
            t = torch.ones(2)
            opts = AllreduceOptions()
            opts.reduceOp = ReduceOp.SUM
            pgs[0].allreduce([t], opts).wait()
            assert t[0].item() == 1.0
        finally:
            for pg in pgs:
                pg.shutdown()


class TestBabyAbort:
    def test_abort_kills_child_and_reconfigure_recovers(self):
        """abort() kills the worker subprocess (aborting any wedged
        collective); a subsequent configure() respawns and works."""
        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
        pg = ProcessGroupBabyGloo(timeout=20.0)
        try:
            pg.configure(f"127.0.0.1:{store.port}/abort0", "r0", 0, 1)
            t = torch.ones(4)
            opts = AllreduceOptions()
            opts.reduceOp = ReduceOp.SUM
            pg.allreduce([t], opts).wait()

            proc = pg._proc
            assert proc is not None and proc.is_alive()
            pg.abort()
            proc.join(timeout=10)
            assert not proc.is_alive()

            # an op against the dead child must fail, not hang
            with pytest.raises(Exception):
                pg.allreduce([torch.ones(4)], opts).wait()

            pg.configure(f"127.0.0.1:{store.port}/abort1", "r0", 0, 1)
            t2 = torch.full((4,), 2.0)
            pg.allreduce([t2], opts).wait()
            torch.testing.assert_close(t2, torch.full((4,), 2.0))
        finally:
            pg.shutdown()
