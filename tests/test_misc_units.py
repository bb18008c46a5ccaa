"""Unit tests for the small support modules: data sharding, stream utils,
pipe helpers, coordination re-exports, and the lighthouse CLI entrypoint.

Reference strategy: torchft's data_test.py / multiprocessing tests — these
cover the same surfaces for the MI355X-native package.
"""

import pickle
import queue
from multiprocessing import Pipe

import pytest
import torch

from torchft_amd.data import DistributedSampler
from torchft_amd.multiprocessing_util import (
    _MonitoredPipe,
    _ThreadPipeEnd,
    _ThreadProcess,
)
from torchft_amd.utils import get_stream_context, record_event, synchronize


class _Dataset(torch.utils.data.Dataset):
    def __init__(self, n: int) -> None:
        self.n = n

    def __len__(self) -> int:
        return self.n

    def __getitem__(self, i: int) -> int:
        return i


class TestDistributedSampler:
    def test_global_rank_layout(self):
        # global rank = group_rank + num_replicas * replica_rank
        s = DistributedSampler(
            _Dataset(96), replica_rank=2, num_replica_groups=3,
            group_rank=1, num_replicas=4,
        )
        assert s.global_rank == 1 + 4 * 2
        assert s.global_world_size == 12
        assert s.rank == s.global_rank
        assert s.num_replicas == s.global_world_size

    def test_shards_are_disjoint_and_cover(self):
        n_groups, n_ranks, n = 2, 2, 64
        seen = []
        for rep in range(n_groups):
            for gr in range(n_ranks):
                s = DistributedSampler(
                    _Dataset(n), replica_rank=rep, num_replica_groups=n_groups,
                    group_rank=gr, num_replicas=n_ranks, shuffle=False,
                )
                seen.append(set(iter(s)))
        # pairwise disjoint, union covers the dataset
        union = set()
        total = 0
        for shard in seen:
            total += len(shard)
            union |= shard
        assert total == n
        assert union == set(range(n))

    def test_requires_ranks_without_dist_init(self):
        # dist not initialized -> group_rank/num_replicas must be explicit
        with pytest.raises(Exception):
            DistributedSampler(_Dataset(8), replica_rank=0, num_replica_groups=1)


class TestStreamUtils:
    def test_null_context_on_cpu(self):
        with get_stream_context(None):
            pass

    def test_record_event_none_on_cpu(self):
        if not torch.cuda.is_available():
            assert record_event() is None

    def test_synchronize_noop_on_cpu(self):
        synchronize()


class TestMonitoredPipe:
    def test_roundtrip_and_timeout(self):
        a, b = Pipe()
        tx, rx = _MonitoredPipe(a), _MonitoredPipe(b)
        tx.send({"x": 1})
        assert rx.recv(timeout=5) == {"x": 1}
        with pytest.raises(TimeoutError):
            rx.recv(timeout=0.05)
        assert not rx.poll(0.0)
        tx.close()
        assert tx.closed()

    def test_exception_rethrown(self):
        a, b = Pipe()
        tx, rx = _MonitoredPipe(a), _MonitoredPipe(b)
        tx.send(ValueError("boom"))
        with pytest.raises(ValueError, match="boom"):
            rx.recv(timeout=5)


class TestThreadPipe:
    def _pair(self):
        q1: "queue.Queue[object]" = queue.Queue()
        q2: "queue.Queue[object]" = queue.Queue()
        return _ThreadPipeEnd(q1, q2), _ThreadPipeEnd(q2, q1)

    def test_send_recv(self):
        a, b = self._pair()
        a.send(41)
        a.send(42)
        assert b.recv() == 41
        assert b.recv() == 42

    def test_poll_is_peek(self):
        a, b = self._pair()
        assert not b.poll(timeout=0.02)
        a.send("msg")
        assert b.poll(timeout=1)
        # poll must not consume
        assert b.recv() == "msg"

    def test_close(self):
        a, _ = self._pair()
        assert not a.closed
        a.close()
        assert a.closed

    def test_thread_process_lifecycle(self):
        ran = []
        p = _ThreadProcess(target=lambda v: ran.append(v), args=(7,))
        assert p.exitcode is None
        p.start()
        p.join(timeout=5)
        assert not p.is_alive()
        assert p.exitcode == 0
        assert ran == [7]
        p.terminate()  # no-op, must not raise
        p.kill()


class TestCoordinationExports:
    def test_reexports(self):
        import torchft_amd.coordination as coord

        for name in coord.__all__:
            assert getattr(coord, name) is not None

    def test_quorum_types_picklable_names(self):
        # the coordination types are C++-backed; they must at least expose
        # usable reprs for logs
        from torchft_amd.coordination import LighthouseServer

        assert "Lighthouse" in LighthouseServer.__name__


class TestLighthouseCLI:
    def test_min_replicas_required(self):
        from torchft_amd.lighthouse import lighthouse_main

        with pytest.raises(SystemExit):
            lighthouse_main([])  # --min_replicas is required

    def test_bad_flag_rejected(self):
        from torchft_amd.lighthouse import lighthouse_main

        with pytest.raises(SystemExit):
            lighthouse_main(["--min_replicas", "1", "--no-such-flag"])


class TestPublicAPI:
    def test_reference_surface_importable(self):
        """The torchft public surface a migrating user needs (docs/MIGRATING.md)
        must stay importable from the package root."""
        import torchft_amd as ft

        for name in [
            "Manager", "WorldSizeMode", "OptimizerWrapper",
            "DistributedDataParallel", "DistributedSampler",
            "ProcessGroup", "ProcessGroupGloo", "ProcessGroupNCCL",
            "ProcessGroupRCCL", "ProcessGroupDummy",
            "ProcessGroupBabyGloo", "ProcessGroupBabyRCCL",
            "ErrorSwallowingProcessGroupWrapper", "ManagedProcessGroup",
            "DiLoCo", "LocalSGD", "split_into_fragments",
        ]:
            assert hasattr(ft, name), f"missing public export: {name}"
            assert name in ft.__all__, f"{name} not in __all__"

    def test_rccl_is_nccl_alias(self):
        # on ROCm the "nccl" backend IS RCCL; both names must resolve to
        # the same class so either spelling works in user code
        import torchft_amd as ft

        assert ft.ProcessGroupRCCL is ft.ProcessGroupNCCL

    def test_submodule_surfaces(self):
        from torchft_amd.checkpointing import HTTPTransport  # noqa: F401
        from torchft_amd.checkpointing.pg_transport import PGTransport  # noqa: F401
        from torchft_amd.collectives import (  # noqa: F401
            allreduce_quantized,
            reduce_scatter_quantized,
        )
        from torchft_amd.coordination import (  # noqa: F401
            LighthouseClient,
            LighthouseServer,
        )
        from torchft_amd.models import LLAMA3_8B, Llama  # noqa: F401
        from torchft_amd.ops import FusedAdamW, RMSNorm  # noqa: F401
        from torchft_amd.parallel.cp import ring_attention  # noqa: F401
