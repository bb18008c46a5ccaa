"""Checkpoint transport tests (reference: torchft/checkpointing/*_test.py)."""

import io
import threading
import time
from datetime import timedelta

import pytest
import torch

from torchft_amd.checkpointing import HTTPTransport, RWLock
from torchft_amd.checkpointing._serialization import streaming_load, streaming_save


class TestSerialization:
    def test_roundtrip_nested(self):
        obj = {
            "model": {
                "w": torch.randn(4, 8),
                "b": torch.arange(10, dtype=torch.int64),
                "half": torch.randn(3, 3).to(torch.bfloat16),
            },
            "step": 7,
            "names": ["a", "b"],
            "empty": torch.empty(0),
            "noncontig": torch.randn(6, 6).t(),
        }
        buf = io.BytesIO()
        streaming_save(obj, buf)
        buf.seek(0)
        loaded = streaming_load(buf)
        assert loaded["step"] == 7
        assert loaded["names"] == ["a", "b"]
        torch.testing.assert_close(loaded["model"]["w"], obj["model"]["w"])
        torch.testing.assert_close(loaded["model"]["b"], obj["model"]["b"])
        torch.testing.assert_close(loaded["model"]["half"], obj["model"]["half"])
        torch.testing.assert_close(loaded["noncontig"], obj["noncontig"].contiguous())
        assert loaded["empty"].numel() == 0

    def test_scalar_and_none_leaves(self):
        obj = {"lr": 0.1, "none": None, "t": torch.ones(2)}
        buf = io.BytesIO()
        streaming_save(obj, buf)
        buf.seek(0)
        loaded = streaming_load(buf)
        assert loaded["lr"] == 0.1
        assert loaded["none"] is None


class TestRWLock:
    def test_concurrent_readers(self):
        lock = RWLock(timeout=5)
        with lock.r_lock():
            with lock.r_lock():
                pass

    def test_writer_excludes_readers(self):
        lock = RWLock(timeout=5)
        lock.w_acquire()
        with pytest.raises(TimeoutError):
            lock.r_acquire(timeout=0.1)
        lock.w_release()
        with lock.r_lock():
            pass

    def test_writer_waits_for_readers(self):
        lock = RWLock(timeout=5)
        lock.r_acquire()
        acquired = threading.Event()

        def writer():
            lock.w_acquire()
            acquired.set()
            lock.w_release()

        t = threading.Thread(target=writer)
        t.start()
        time.sleep(0.1)
        assert not acquired.is_set()
        lock.r_release()
        t.join(timeout=5)
        assert acquired.is_set()


class TestHTTPTransport:
    def test_send_recv(self):
        src = HTTPTransport(timeout=timedelta(seconds=10))
        dst = HTTPTransport(timeout=timedelta(seconds=10))
        try:
            sd = {"user": {"default": {"w": torch.randn(16, 16)}}, "torchft": {"step": 3}}
            src.send_checkpoint([1], step=3, state_dict=sd, timeout=timedelta(seconds=10))
            got = dst.recv_checkpoint(
                src_rank=0, metadata=src.metadata(), step=3, timeout=timedelta(seconds=10)
            )
            torch.testing.assert_close(got["user"]["default"]["w"], sd["user"]["default"]["w"])
            assert got["torchft"]["step"] == 3
        finally:
            src.shutdown()
            dst.shutdown()

    def test_wrong_step_rejected(self):
        src = HTTPTransport(timeout=timedelta(seconds=10))
        dst = HTTPTransport(timeout=timedelta(seconds=10))
        try:
            src.send_checkpoint(
                [1], step=3, state_dict={"x": 1}, timeout=timedelta(seconds=10)
            )
            with pytest.raises(Exception):
                dst.recv_checkpoint(
                    src_rank=0,
                    metadata=src.metadata(),
                    step=99,
                    timeout=timedelta(seconds=5),
                )
        finally:
            src.shutdown()
            dst.shutdown()

    def test_snapshot_survives_disallow(self):
        # The staged checkpoint is a snapshot, so (unlike the reference's
        # live-serving transport) it stays fetchable after training resumes;
        # this removes the slow-fetcher vs next-step race.
        src = HTTPTransport(timeout=timedelta(seconds=10))
        dst = HTTPTransport(timeout=timedelta(seconds=10))
        try:
            t = torch.ones(4)
            src.send_checkpoint(
                [1], step=3, state_dict={"x": t}, timeout=timedelta(seconds=10)
            )
            t.mul_(100.0)  # training resumed and mutated the weights
            src.disallow_checkpoint()
            got = dst.recv_checkpoint(
                src_rank=0, metadata=src.metadata(), step=3,
                timeout=timedelta(seconds=5),
            )
            torch.testing.assert_close(got["x"], torch.ones(4))  # snapshot value
        finally:
            src.shutdown()
            dst.shutdown()

    def test_chunked_fetch(self):
        src = HTTPTransport(timeout=timedelta(seconds=10), num_chunks=3)
        dst = HTTPTransport(timeout=timedelta(seconds=10), num_chunks=3)
        try:
            sd = {f"t{i}": torch.randn(32) for i in range(7)}
            sd["meta"] = "hello"
            src.send_checkpoint([1], step=1, state_dict=sd, timeout=timedelta(seconds=10))
            got = dst.recv_checkpoint(
                src_rank=0, metadata=src.metadata(), step=1, timeout=timedelta(seconds=10)
            )
            for i in range(7):
                torch.testing.assert_close(got[f"t{i}"], sd[f"t{i}"])
            assert got["meta"] == "hello"
        finally:
            src.shutdown()
            dst.shutdown()


class TestHTTPTransportErrorPaths:
    def test_unreachable_source_raises_promptly(self):
        import socket
        import time
        import urllib.error

        # grab a port nothing listens on
        with socket.socket() as s:
            s.bind(("127.0.0.1", 0))
            dead_port = s.getsockname()[1]

        dst = HTTPTransport(timeout=timedelta(seconds=5))
        try:
            t0 = time.monotonic()
            with pytest.raises((urllib.error.URLError, OSError)):
                dst.recv_checkpoint(
                    src_rank=0,
                    metadata=f"http://127.0.0.1:{dead_port}",
                    step=1,
                    timeout=timedelta(seconds=5),
                )
            assert time.monotonic() - t0 < 5.5  # bounded, not hung
        finally:
            dst.shutdown()

    def test_fetch_retries_until_staged(self):
        # the destination may issue its GET a beat before the source's
        # quorum thread stages the snapshot; _open_with_retry absorbs the
        # 400 window instead of failing the heal
        import threading

        src = HTTPTransport(timeout=timedelta(seconds=10))
        dst = HTTPTransport(timeout=timedelta(seconds=10))
        try:
            sd = {"w": torch.arange(8.0)}

            def stage_late():
                threading.Event().wait(0.4)
                src.send_checkpoint(
                    [1], step=7, state_dict=sd, timeout=timedelta(seconds=10)
                )

            t = threading.Thread(target=stage_late)
            t.start()
            got = dst.recv_checkpoint(
                src_rank=0, metadata=src.metadata(), step=7,
                timeout=timedelta(seconds=10),
            )
            t.join()
            torch.testing.assert_close(got["w"], sd["w"])
        finally:
            src.shutdown()
            dst.shutdown()


class TestDTensorSerialization:
    def test_dtensor_streaming_roundtrip(self):
        """FSDP2 sharded state dicts hold DTensors; the streaming format
        carries (local shard + pickled spec) and reconstructs the DTensor
        (the HSDP heal path)."""
        import io
        import os

        import torch.distributed as dist
        from torch.distributed.device_mesh import init_device_mesh
        from torch.distributed.tensor import DTensor, Shard, distribute_tensor

        from torchft_amd.checkpointing._serialization import (
            streaming_load,
            streaming_save,
        )

        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29801")
        created = not dist.is_initialized()
        if created:
            dist.init_process_group("gloo", rank=0, world_size=1)
        try:
            mesh = init_device_mesh("cpu", (1,))
            full = torch.randn(8, 4)
            dt = distribute_tensor(full, mesh, [Shard(0)])

            buf = io.BytesIO()
            streaming_save({"w": dt, "plain": torch.arange(4.0), "meta": 3}, buf)
            buf.seek(0)
            out = streaming_load(buf)
            assert isinstance(out["w"], DTensor)
            torch.testing.assert_close(out["w"].to_local(), dt.to_local())
            torch.testing.assert_close(out["plain"], torch.arange(4.0))
            assert out["meta"] == 3
        finally:
            if created:
                dist.destroy_process_group()


class TestChunkedDTensorFetch:
    def test_chunked_transport_carries_dtensor(self):
        """num_chunks>0 fetches tensor payloads over parallel range
        requests; DTensor leaves must reassemble with their spec intact."""
        import os

        import torch.distributed as dist
        from torch.distributed.device_mesh import init_device_mesh
        from torch.distributed.tensor import DTensor, Shard, distribute_tensor

        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29801")
        created = not dist.is_initialized()
        if created:
            dist.init_process_group("gloo", rank=0, world_size=1)
        src = HTTPTransport(timeout=timedelta(seconds=10), num_chunks=2)
        dst = HTTPTransport(timeout=timedelta(seconds=10), num_chunks=2)
        try:
            mesh = init_device_mesh("cpu", (1,))
            dt = distribute_tensor(torch.randn(16, 4), mesh, [Shard(0)])
            sd = {"sharded": dt, "plain": torch.randn(8), "step": 5}
            src.send_checkpoint([1], step=5, state_dict=sd,
                                timeout=timedelta(seconds=10))
            got = dst.recv_checkpoint(
                src_rank=0, metadata=src.metadata(), step=5,
                timeout=timedelta(seconds=10),
            )
            assert isinstance(got["sharded"], DTensor)
            torch.testing.assert_close(got["sharded"].to_local(), dt.to_local())
            torch.testing.assert_close(got["plain"], sd["plain"])
            assert got["step"] == 5
        finally:
            src.shutdown()
            dst.shutdown()
            if created:
                dist.destroy_process_group()
