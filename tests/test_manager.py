"""Manager unit tests with a mocked coordination client.

Reference strategy: torchft/manager_test.py — drive quorum/heal/commit
paths with a MagicMock ManagerClient and ProcessGroupDummy, no servers.
"""

from datetime import timedelta
from typing import Optional
from unittest.mock import MagicMock, patch

import pytest
import torch
from torch.distributed import TCPStore

from torchft_amd._ftcore import QuorumResult
from torchft_amd.manager import Manager, WorldSizeMode
from torchft_amd.process_group import ProcessGroupDummy


def mock_quorum(
    quorum_id=1,
    replica_rank=0,
    replica_world_size=2,
    recover_src_manager_address="",
    recover_src_replica_rank: Optional[int] = None,
    recover_dst_replica_ranks=(),
    store_address="127.0.0.1:0",
    max_step=0,
    max_replica_rank: Optional[int] = 0,
    max_world_size=2,
    heal=False,
    commit_failures=0,
    replica_ids=("a", "b"),
) -> QuorumResult:
    q = QuorumResult()
    q.quorum_id = quorum_id
    q.replica_rank = replica_rank
    q.replica_world_size = replica_world_size
    q.recover_src_manager_address = recover_src_manager_address
    q.recover_src_replica_rank = recover_src_replica_rank
    q.recover_dst_replica_ranks = list(recover_dst_replica_ranks)
    q.store_address = store_address
    q.max_step = max_step
    q.max_replica_rank = max_replica_rank
    q.max_world_size = max_world_size
    q.heal = heal
    q.commit_failures = commit_failures
    q.replica_ids = list(replica_ids)
    return q


def make_manager(client: MagicMock, use_async_quorum=True, **kwargs) -> Manager:
    store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
    state = {"w": torch.zeros(4)}
    with patch("torchft_amd.manager.ManagerClient", return_value=client), patch(
        "torchft_amd.manager.ManagerServer"
    ) as mock_server:
        mock_server.return_value.address.return_value = "http://127.0.0.1:1"
        manager = Manager(
            pg=ProcessGroupDummy(0, 1),
            load_state_dict=lambda sd: state.update(sd),
            state_dict=lambda: state,
            min_replica_size=1,
            use_async_quorum=use_async_quorum,
            rank=0,
            world_size=1,
            store_addr="127.0.0.1",
            store_port=store.port,
            lighthouse_addr="http://nowhere:1",
            replica_id="test1",
            hostname="127.0.0.1",
            timeout=timedelta(seconds=5),
            **kwargs,
        )
    manager._test_store = store  # keep the store alive
    manager._test_state = state
    return manager


class TestManagerQuorum:
    def test_healthy_step_commits(self):
        client = MagicMock()
        client._quorum.return_value = mock_quorum()
        client.should_commit.return_value = True
        m = make_manager(client)
        try:
            m.start_quorum()
            t = torch.ones(4)
            work = m.allreduce(t)
            assert work.wait()
            # AVG normalization by participants (max_world_size=2)
            torch.testing.assert_close(t, torch.full((4,), 0.5))
            assert m.should_commit()
            assert m.current_step() == 1
            assert m.batches_committed() == 2
        finally:
            m.shutdown(wait=False)

    def test_quorum_id_change_reconfigures_pg(self):
        client = MagicMock()
        client._quorum.return_value = mock_quorum(quorum_id=5)
        client.should_commit.return_value = True
        m = make_manager(client)
        try:
            m.start_quorum()
            m.wait_quorum()
            assert m._pg.configure_count == 1
            # same quorum id -> no reconfigure
            m.should_commit()
            m.start_quorum()
            m.wait_quorum()
            assert m._pg.configure_count == 1
            # new quorum id -> reconfigure
            client._quorum.return_value = mock_quorum(quorum_id=6, max_step=1)
            m.should_commit()
            m.start_quorum()
            m.wait_quorum()
            assert m._pg.configure_count == 2
        finally:
            m.shutdown(wait=False)

    def test_error_zeros_and_rejects_commit(self):
        client = MagicMock()
        client._quorum.return_value = mock_quorum()
        client.should_commit.return_value = False
        m = make_manager(client)
        try:
            m.start_quorum()
            m.wait_quorum()
            m.report_error(RuntimeError("boom"))
            t = torch.ones(4)
            work = m.allreduce(t)  # becomes a no-op after error
            assert work.wait()
            assert not m.should_commit()
            assert m.current_step() == 0
            # client is told should_commit=False
            assert client.should_commit.call_args[0][2] is False
        finally:
            m.shutdown(wait=False)

    def test_healing_async_not_participating(self):
        client = MagicMock()
        client._quorum.return_value = mock_quorum(
            replica_rank=1,
            recover_src_manager_address="http://src:1",
            recover_src_replica_rank=0,
            max_step=5,
            max_replica_rank=None,
            max_world_size=1,
            heal=True,
        )
        client.should_commit.return_value = True
        src_client = MagicMock()
        src_client._checkpoint_metadata.return_value = "http://src-ckpt:1"

        m = make_manager(client)
        try:
            transport = MagicMock()
            transport.metadata.return_value = "http://me:1"
            transport.recv_checkpoint.return_value = {
                "user": {"default": {"w": torch.full((4,), 7.0)}},
                "torchft": {"step": 5, "batches_committed": 10},
            }
            m._checkpoint_transport = transport
            with patch("torchft_amd.manager.ManagerClient", return_value=src_client):
                m.start_quorum()
                m.wait_quorum()
            assert m._healing
            assert not m.is_participating()
            assert m.num_participants() == 1
            t = torch.ones(4)
            m.allreduce(t).wait()
            # non-participating grads are zeroed
            torch.testing.assert_close(t, torch.zeros(4))
            assert m.should_commit()
            # healed state applied on the main thread at should_commit
            torch.testing.assert_close(m._test_state["w"], torch.full((4,), 7.0))
            assert m.current_step() == 6  # healed to 5, committed one step
        finally:
            m.shutdown(wait=False)

    def test_sync_quorum_applies_state_eagerly(self):
        client = MagicMock()
        client._quorum.return_value = mock_quorum(
            replica_rank=1,
            recover_src_manager_address="http://src:1",
            recover_src_replica_rank=0,
            max_step=3,
            max_replica_rank=None,
            heal=True,
        )
        src_client = MagicMock()
        src_client._checkpoint_metadata.return_value = "meta"
        m = make_manager(client, use_async_quorum=False)
        try:
            transport = MagicMock()
            transport.metadata.return_value = "m"
            transport.recv_checkpoint.return_value = {
                "user": {"default": {"w": torch.full((4,), 3.0)}},
                "torchft": {"step": 3, "batches_committed": 6},
            }
            m._checkpoint_transport = transport
            with patch("torchft_amd.manager.ManagerClient", return_value=src_client):
                m.start_quorum()
            # state applied eagerly; participating in sync mode
            torch.testing.assert_close(m._test_state["w"], torch.full((4,), 3.0))
            assert not m._healing
            assert m.is_participating()
            assert m.current_step() == 3
        finally:
            m.shutdown(wait=False)

    def test_fixed_with_spares_caps_world(self):
        client = MagicMock()
        client._quorum.return_value = mock_quorum(
            replica_rank=1, max_replica_rank=1, max_world_size=2, replica_world_size=2
        )
        m = make_manager(
            client,
            world_size_mode=WorldSizeMode.FIXED_WITH_SPARES,
        )
        # min_replica_size=1 -> only 1 participates; rank 1 is a spare
        try:
            m.start_quorum()
            m.wait_quorum()
            assert m.num_participants() == 1
            assert m.participating_rank() is None
        finally:
            m.shutdown(wait=False)

    def test_max_retries_raises(self):
        client = MagicMock()
        client._quorum.return_value = mock_quorum()
        client.should_commit.return_value = False
        m = make_manager(client, max_retries=1)
        try:
            m.start_quorum()
            assert not m.should_commit()  # failure 1
            m.start_quorum()
            with pytest.raises(RuntimeError, match="max_retries"):
                m.should_commit()  # failure 2 > max_retries
        finally:
            m.shutdown(wait=False)

    def test_commit_failures_forwarded_to_quorum(self):
        client = MagicMock()
        client._quorum.return_value = mock_quorum()
        client.should_commit.return_value = False
        m = make_manager(client)
        try:
            m.start_quorum()
            m.should_commit()
            m.start_quorum()
            m.wait_quorum()
            assert client._quorum.call_args.kwargs["commit_failures"] == 1
        finally:
            m.shutdown(wait=False)

    def test_report_error_and_wrap_future(self):
        client = MagicMock()
        client._quorum.return_value = mock_quorum()
        m = make_manager(client)
        try:
            fut: torch.futures.Future = torch.futures.Future()
            wrapped = m.wrap_future(fut, default=torch.zeros(2))
            fut.set_exception(RuntimeError("inner fail"))
            out = wrapped.wait()
            torch.testing.assert_close(out, torch.zeros(2))
            assert m.errored() is not None
        finally:
            m.shutdown(wait=False)


class TestManagedWorkChain:
    def test_lazy_then_chain_materializes_on_wait(self):
        from torchft_amd.manager import _ManagedWork
        from torchft_amd.work import _DummyWork

        client = MagicMock()
        client._quorum.return_value = mock_quorum()
        m = make_manager(client)
        try:
            m.start_quorum()
            m.wait_quorum()
            t = torch.ones(4)
            inner = _DummyWork([t])
            work = _ManagedWork(m, inner, t)
            fut = work.get_future()
            calls = []

            def cb(f):
                calls.append(1)
                return f.value() * 2

            fut = fut.then(cb)
            # then() is lazy: nothing ran yet
            assert calls == []
            assert work.wait()
            assert calls == [1]
            torch.testing.assert_close(fut.wait(), torch.full((4,), 2.0))
        finally:
            m.shutdown(wait=False)

    def test_allreduce_sum_skips_normalization(self):
        from torch.distributed import ReduceOp

        client = MagicMock()
        client._quorum.return_value = mock_quorum()
        m = make_manager(client)
        try:
            m.start_quorum()
            t = torch.ones(4)
            work = m.allreduce(t, reduce_op=ReduceOp.SUM)
            assert work.wait()
            torch.testing.assert_close(t, torch.ones(4))  # no /participants
        finally:
            m.shutdown(wait=False)

    def test_avg_rejects_integer_tensors(self):
        client = MagicMock()
        client._quorum.return_value = mock_quorum()
        m = make_manager(client)
        try:
            m.start_quorum()
            with pytest.raises(ValueError, match="floating point"):
                m.allreduce(torch.ones(4, dtype=torch.int64))
        finally:
            m.shutdown(wait=False)


class TestQuorumArgsPropagation:
    def test_shrink_only_forwarded(self):
        client = MagicMock()
        client._quorum.return_value = mock_quorum()
        m = make_manager(client)
        try:
            m.start_quorum(shrink_only=True)
            m.wait_quorum()
            assert client._quorum.call_args.kwargs["shrink_only"] is True
            assert client._quorum.call_args.kwargs["init_sync"] is True
        finally:
            m.shutdown(wait=False)


class TestManagerCounters:
    """Accessors and counter persistence (reference: manager_test.py's
    state_dict/load_state_dict and participation accessor coverage)."""

    def test_accessors_before_any_quorum(self):
        client = MagicMock()
        m = make_manager(client)
        try:
            assert m.participating_rank() is None
            assert m.num_participants() == 0
            assert not m.is_participating()
            assert m.current_step() == 0
            assert m.batches_committed() == 0
        finally:
            m.shutdown(wait=False)

    def test_counters_roundtrip(self):
        client = MagicMock()
        m = make_manager(client)
        try:
            m.load_state_dict({"step": 7, "batches_committed": 21})
            assert m.current_step() == 7
            assert m.batches_committed() == 21
            assert m.state_dict() == {"step": 7, "batches_committed": 21}
        finally:
            m.shutdown(wait=False)

    def test_batches_committed_accumulates_participants(self):
        # every committed step adds num_participants (the global batch is
        # sized by live replicas, not a constant)
        client = MagicMock()
        client._quorum.return_value = mock_quorum()
        client.should_commit.return_value = True
        m = make_manager(client)
        try:
            for expected in (2, 4):
                m.start_quorum()
                m.allreduce(torch.ones(4)).wait()
                assert m.num_participants() == 2
                assert m.should_commit()
                assert m.batches_committed() == expected
                client._quorum.return_value = mock_quorum(
                    max_step=m.current_step()
                )
        finally:
            m.shutdown(wait=False)

    def test_failed_commit_does_not_advance_counters(self):
        client = MagicMock()
        client._quorum.return_value = mock_quorum()
        client.should_commit.return_value = False
        m = make_manager(client)
        try:
            m.start_quorum()
            m.allreduce(torch.ones(4)).wait()
            assert not m.should_commit()
            assert m.current_step() == 0
            assert m.batches_committed() == 0
        finally:
            m.shutdown(wait=False)


class TestOptimizerWrapper:
    """zero_grad starts the quorum; step is gated by should_commit
    (reference: torchft/optim.py semantics)."""

    def _wrapped(self, client):
        from torchft_amd.optim import OptimizerWrapper

        m = make_manager(client)
        lin = torch.nn.Linear(4, 2)
        base = torch.optim.SGD(lin.parameters(), lr=0.1)
        return m, lin, base, OptimizerWrapper(m, base)

    def test_step_applies_only_on_commit(self):
        client = MagicMock()
        client._quorum.return_value = mock_quorum()
        client.should_commit.return_value = False
        m, lin, base, opt = self._wrapped(client)
        try:
            before = {k: v.clone() for k, v in lin.state_dict().items()}
            opt.zero_grad()
            lin(torch.randn(3, 4)).sum().backward()
            opt.step()  # vote fails -> parameters must not move
            for k, v in lin.state_dict().items():
                torch.testing.assert_close(v, before[k], rtol=0, atol=0)

            client.should_commit.return_value = True
            opt.zero_grad()
            lin(torch.randn(3, 4)).sum().backward()
            opt.step()  # commit -> parameters move
            moved = any(
                not torch.equal(v, before[k]) for k, v in lin.state_dict().items()
            )
            assert moved
        finally:
            m.shutdown(wait=False)

    def test_zero_grad_starts_quorum(self):
        client = MagicMock()
        client._quorum.return_value = mock_quorum()
        m, lin, base, opt = self._wrapped(client)
        try:
            assert m.participating_rank() is None
            opt.zero_grad()
            m.wait_quorum()
            assert client._quorum.called
        finally:
            m.shutdown(wait=False)

    def test_state_dict_passthrough(self):
        client = MagicMock()
        m, lin, base, opt = self._wrapped(client)
        try:
            assert opt.state_dict() == base.state_dict()
            opt.load_state_dict(base.state_dict())
            assert opt.param_groups is base.param_groups
            assert opt.state is base.state
            with pytest.raises(AssertionError):
                opt.step(closure=lambda: 0.0)
        finally:
            m.shutdown(wait=False)


class TestPureDDP:
    def test_grads_averaged_like_reducer_ddp(self):
        """PureDDP is the per-parameter correctness oracle: its grads must
        match the reducer-based DDP wrapper bitwise (both AVG over the
        quorum's participant count)."""
        from torchft_amd.ddp import (
            DistributedDataParallel,
            PureDistributedDataParallel,
        )

        def run(wrapper_cls):
            client = MagicMock()
            client._quorum.return_value = mock_quorum()  # 2 participants
            client.should_commit.return_value = True
            m = make_manager(client)
            try:
                torch.manual_seed(7)
                model = torch.nn.Sequential(
                    torch.nn.Linear(6, 8), torch.nn.ReLU(), torch.nn.Linear(8, 3)
                )
                wrapped = wrapper_cls(m, model)
                m.start_quorum()
                gen = torch.Generator().manual_seed(11)
                x = torch.randn(4, 6, generator=gen)
                wrapped(x).sum().backward()
                assert m.should_commit()
                return [p.grad.clone() for p in model.parameters()]
            finally:
                m.shutdown(wait=False)

        pure = run(PureDistributedDataParallel)
        reducer = run(DistributedDataParallel)
        for gp, gr in zip(pure, reducer):
            torch.testing.assert_close(gp, gr, rtol=0, atol=0)


class TestQuantizedFallback:
    def test_should_quantize_falls_back_on_cpu(self):
        # the fp8 path needs a HIP device; on CPU hosts the managed
        # allreduce must silently use the plain wire path, not raise
        client = MagicMock()
        client._quorum.return_value = mock_quorum()
        client.should_commit.return_value = True
        m = make_manager(client, should_quantize=True)
        try:
            m.start_quorum()
            t = torch.ones(4)
            assert m.allreduce(t).wait()
            torch.testing.assert_close(t, torch.full((4,), 0.5))  # AVG by 2
            assert m.should_commit()
        finally:
            m.shutdown(wait=False)


class TestManagerLifecycleHygiene:
    def test_repeated_construct_shutdown_no_fd_growth(self):
        """Manager owns sockets, a serial-executor thread, and a checkpoint
        transport; repeated lifecycle must not accumulate fds or threads."""
        import gc
        import os
        import threading

        def nfds():
            return len(os.listdir(f"/proc/{os.getpid()}/fd"))

        # warm-up cycle so lazily created module state is excluded
        for i in range(2):
            m = make_manager(MagicMock())
            m.shutdown(wait=True)
        gc.collect()
        base_fds, base_thr = nfds(), threading.active_count()
        for i in range(8):
            m = make_manager(MagicMock())
            m.shutdown(wait=True)
            del m
        gc.collect()
        assert nfds() <= base_fds + 4, "fd growth across manager lifecycles"
        assert threading.active_count() <= base_thr + 1


class TestGlobalRankDerivation:
    def test_trailing_digits(self):
        from torchft_amd.manager import extract_trailing_digits

        assert extract_trailing_digits("replica_57") == 57
        assert extract_trailing_digits("bench0") == 0
        assert extract_trailing_digits("train_ddp_12") == 12
        assert extract_trailing_digits("no_digits_") == 0
        assert extract_trailing_digits("") == 0
        assert extract_trailing_digits("123") == 123

    def test_global_rank_formula(self):
        # global_rank = replica_index * group_world_size + group_rank
        # (reference: manager.py's trailing-digit derivation)
        client = MagicMock()
        m = make_manager(client)  # harness replica_id is "test1" -> index 1
        try:
            assert m._global_rank == 1 * 1 + 0
        finally:
            m.shutdown(wait=False)


class TestManagedProcessGroup:
    def test_routes_allreduce_and_size(self):
        """ManagedProcessGroup adapts a Manager to the PG interface so
        stock DDP/FSDP hooks get the FT allreduce; size() is the live
        participant count (reference: ManagedProcessGroup semantics)."""
        from torch.distributed import ReduceOp as RO
        from torch.distributed.distributed_c10d import AllreduceOptions

        from torchft_amd.process_group import ManagedProcessGroup

        client = MagicMock()
        client._quorum.return_value = mock_quorum()  # 2 participants
        client.should_commit.return_value = True
        m = make_manager(client)
        try:
            mpg = ManagedProcessGroup(m)
            m.start_quorum()
            assert mpg.size() == 2

            t = torch.ones(4)
            opts = AllreduceOptions()
            opts.reduceOp = RO.AVG
            assert mpg.allreduce([t], opts).wait()
            torch.testing.assert_close(t, torch.full((4,), 0.5))  # AVG by 2

            t2 = torch.ones(4)
            assert mpg.allreduce([t2], RO.SUM).wait()  # bare-op form
            torch.testing.assert_close(t2, torch.ones(4))

            with pytest.raises(AssertionError):
                mpg.allreduce([t, t2], opts)  # exactly one tensor
            assert m.should_commit()
        finally:
            m.shutdown(wait=False)
