"""Fault-path resiliency tests.

Reference strategy: torchft/process_group_test.py `_run_with_resiliency` —
a peer dies mid-collective; survivors surface an error (or timeout), then
``configure()`` a smaller world and the collective succeeds again. Plus a
TP-composition smoke (reference: fsdp_test.py test_fsdp_tp) showing tensor
parallelism composes with the managed FT dimension.
"""

import threading
import time
from concurrent.futures import ThreadPoolExecutor
from datetime import timedelta
from unittest.mock import MagicMock

import pytest
import torch
import torch.nn as nn
from torch.distributed import TCPStore
from torch.distributed.distributed_c10d import AllreduceOptions, ReduceOp

from torchft_amd.process_group import ProcessGroupGloo


class TestResiliency:
    def test_peer_death_then_shrink_and_recover(self):
        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
        barrier = threading.Barrier(2)

        def survivor():
            pg = ProcessGroupGloo(timeout=timedelta(seconds=3))
            pg.configure(f"127.0.0.1:{store.port}/q0", "r0", 0, 2)
            barrier.wait()
            # the peer dies without participating: this op must error or
            # time out rather than hang forever
            t = torch.ones(4)
            opts = AllreduceOptions()
            opts.reduceOp = ReduceOp.SUM
            with pytest.raises(Exception):
                work = pg.allreduce([t], opts)
                work.wait()
                # gloo may only detect on next op
                pg.allreduce([t], opts).wait()
                raise RuntimeError("collective with dead peer succeeded twice")
            # reconfigure to a world of 1 and carry on
            pg.configure(f"127.0.0.1:{store.port}/q1", "r0", 0, 1)
            t2 = torch.ones(4)
            pg.allreduce([t2], opts).wait()
            torch.testing.assert_close(t2, torch.ones(4))
            return True

        def dying_peer():
            pg = ProcessGroupGloo(timeout=timedelta(seconds=3))
            pg.configure(f"127.0.0.1:{store.port}/q0", "r1", 1, 2)
            barrier.wait()
            time.sleep(0.2)
            pg.abort()  # dies without issuing the collective
            return True

        with ThreadPoolExecutor(max_workers=2) as ex:
            fs = [ex.submit(survivor), ex.submit(dying_peer)]
            assert all(f.result(timeout=60) for f in fs)

    def test_abort_makes_collectives_fail_fast(self):
        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
        pg = ProcessGroupGloo(timeout=timedelta(seconds=2))
        pg.configure(f"127.0.0.1:{store.port}/a", "r0", 0, 1)
        pg.abort()
        t = torch.ones(2)
        opts = AllreduceOptions()
        opts.reduceOp = ReduceOp.SUM
        with pytest.raises(Exception):
            pg.allreduce([t], opts).wait()


class TestTPComposition:
    def test_tp_with_managed_allreduce(self):
        """ColwiseParallel over a CPU mesh + the managed cross-replica
        allreduce (mocked manager) — the two dimensions are orthogonal."""
        import os

        import torch.distributed as dist
        from torch.distributed.tensor import init_device_mesh
        from torch.distributed.tensor.parallel import (
            ColwiseParallel,
            parallelize_module,
        )

        from torchft_amd.manager import Manager
        from torchft_amd.process_group import ManagedProcessGroup, ProcessGroupDummy

        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
        dist.init_process_group(
            "gloo", store=dist.PrefixStore("tp", store), rank=0, world_size=1
        )
        try:
            mesh = init_device_mesh("cpu", (1,), mesh_dim_names=("tp",))
            model = nn.Sequential(nn.Linear(16, 32), nn.ReLU(), nn.Linear(32, 16))
            model = parallelize_module(model, mesh, {"0": ColwiseParallel()})

            manager = MagicMock(spec=Manager)
            manager._pg = ProcessGroupDummy(0, 1)
            manager.num_participants.return_value = 1

            from torchft_amd.work import _DummyWork

            def managed_allreduce(tensor, **kwargs):
                return _DummyWork(tensor)

            manager.allreduce.side_effect = managed_allreduce
            mpg = ManagedProcessGroup(manager)

            out = model(torch.randn(4, 16))
            loss = out.sum()
            loss.backward()

            # cross-replica allreduce of (sharded) grads through the manager
            for p in model.parameters():
                g = p.grad
                if g is None:
                    continue
                local = g.to_local() if hasattr(g, "to_local") else g
                opts = AllreduceOptions()
                opts.reduceOp = ReduceOp.SUM
                mpg.allreduce([local], opts).wait()
            assert manager.allreduce.call_count > 0
        finally:
            dist.destroy_process_group()
