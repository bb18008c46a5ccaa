"""LocalSGD / DiLoCo integration tests (threads-as-replicas over gloo).

Reference strategy: torchft/local_sgd_integ_test.py — healthy runs and
recovery, asserting the global (outer) state matches across replicas.
"""

import logging
import threading
from concurrent.futures import ThreadPoolExecutor
from datetime import timedelta
from typing import Dict, List

import pytest
import torch
import torch.nn as nn
from torch.distributed import TCPStore

from torchft_amd._ftcore import LighthouseServer
from torchft_amd.local_sgd import DiLoCo, LocalSGD
from torchft_amd.manager import Manager
from torchft_amd.process_group import ProcessGroupGloo

logging.getLogger("torchft_amd").setLevel(logging.WARNING)


def _make_model() -> nn.Module:
    torch.manual_seed(7)
    return nn.Sequential(nn.Linear(4, 8), nn.ReLU(), nn.Linear(8, 4))


class InjectedFailure(Exception):
    pass


def _localsgd_replica(replica_id: int, lighthouse_addr: str, total_outer: int,
                      sync_every: int) -> Dict[str, torch.Tensor]:
    store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
    model = _make_model()
    manager = Manager(
        pg=ProcessGroupGloo(timeout=timedelta(seconds=20)),
        load_state_dict=model.load_state_dict,
        state_dict=model.state_dict,
        min_replica_size=1,
        rank=0,
        world_size=1,
        store_addr="127.0.0.1",
        store_port=store.port,
        lighthouse_addr=lighthouse_addr,
        replica_id=f"ls_{replica_id}",
        hostname="127.0.0.1",
        timeout=timedelta(seconds=20),
    )
    try:
        opt = torch.optim.SGD(model.parameters(), lr=0.05)
        criterion = nn.MSELoss()
        with LocalSGD(manager, model, opt, sync_every=sync_every):
            while manager.current_step() < total_outer:
                # replica-dependent data: local models diverge between syncs
                torch.manual_seed(manager.current_step() * 10 + replica_id)
                x = torch.randn(4, 4)
                y = torch.randn(4, 4)
                opt.zero_grad()
                criterion(model(x), y).backward()
                opt.step()
        return {k: v.detach().clone() for k, v in model.state_dict().items()}
    finally:
        manager.shutdown(wait=False)


class TestLocalSGD:
    def test_two_replicas_converge(self):
        lh = LighthouseServer(bind="127.0.0.1:0", min_replicas=2, join_timeout_ms=1000)
        try:
            with ThreadPoolExecutor(max_workers=2) as ex:
                futs = [
                    ex.submit(_localsgd_replica, i, lh.address(), 3, 2)
                    for i in range(2)
                ]
                dicts = [f.result(timeout=60) for f in futs]
            for k in dicts[0]:
                torch.testing.assert_close(dicts[0][k], dicts[1][k])
        finally:
            lh.shutdown()


def _diloco_replica(
    replica_id: int,
    lighthouse_addr: str,
    total_outer: int,
    sync_every: int,
    fail_at_step: int = -1,
    attempts: int = 3,
) -> Dict[str, object]:
    did_fail = False
    for attempt in range(attempts):
        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
        model = _make_model()
        fragments = [model[0], model[2]]
        inner_opt = torch.optim.SGD(model.parameters(), lr=0.05)
        # one outer optimizer per fragment, over that fragment's params only
        # (a shared outer optimizer would also step the non-synced fragment
        # with stale inner grads)
        outer_opt = [
            torch.optim.SGD(f.parameters(), lr=0.5, momentum=0.9) for f in fragments
        ]
        manager = Manager(
            pg=ProcessGroupGloo(timeout=timedelta(seconds=20)),
            load_state_dict=model.load_state_dict,
            state_dict=model.state_dict,
            min_replica_size=1,
            use_async_quorum=False,
            rank=0,
            world_size=1,
            store_addr="127.0.0.1",
            store_port=store.port,
            lighthouse_addr=lighthouse_addr,
            replica_id=f"dl_{replica_id}",
            hostname="127.0.0.1",
            timeout=timedelta(seconds=20),
        )
        criterion = nn.MSELoss()
        fail_armed = fail_at_step >= 0 and attempt == 0
        try:
            diloco = DiLoCo(
                manager,
                fragments,
                inner_opt,
                outer_opt,
                sync_every=sync_every,
                pin_memory=False,
            )
            with diloco:
                local_batches = 0
                while manager.current_step() < total_outer:
                    if fail_armed and manager.current_step() >= fail_at_step:
                        raise InjectedFailure(f"fail replica {replica_id}")
                    torch.manual_seed(local_batches * 10 + replica_id)
                    x = torch.randn(4, 4)
                    y = torch.randn(4, 4)
                    inner_opt.zero_grad()
                    criterion(model(x), y).backward()
                    inner_opt.step()
                    local_batches += 1
                    # the DiLoCo post-hook may have live-healed PAST the
                    # target step (slow start under suite load); fire here
                    # too or the loop exits with the failure never injected
                    if fail_armed and manager.current_step() >= fail_at_step:
                        raise InjectedFailure(f"fail replica {replica_id}")
            return {
                "original": {
                    f"{i}_{k}": v.detach().clone()
                    for i, frag in enumerate(diloco._fragments)
                    for k, v in frag.original_parameters.items()
                },
                "failed": did_fail,
            }
        except InjectedFailure:
            did_fail = True
            continue
        finally:
            manager.shutdown(wait=False)
    raise RuntimeError("unreachable")


class TestDiLoCo:
    def test_two_replicas_healthy(self):
        lh = LighthouseServer(bind="127.0.0.1:0", min_replicas=2, join_timeout_ms=1000)
        try:
            with ThreadPoolExecutor(max_workers=2) as ex:
                futs = [
                    ex.submit(_diloco_replica, i, lh.address(), 2, 4)
                    for i in range(2)
                ]
                results = [f.result(timeout=90) for f in futs]
            a, b = results[0]["original"], results[1]["original"]
            for k in a:
                torch.testing.assert_close(a[k], b[k], msg=f"mismatch at {k}")
        finally:
            lh.shutdown()

    def test_validation_rejects_async_quorum(self):
        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
        lh = LighthouseServer(bind="127.0.0.1:0", min_replicas=1, join_timeout_ms=50)
        model = _make_model()
        manager = Manager(
            pg=ProcessGroupGloo(),
            load_state_dict=model.load_state_dict,
            state_dict=model.state_dict,
            min_replica_size=1,
            use_async_quorum=True,
            rank=0,
            world_size=1,
            store_addr="127.0.0.1",
            store_port=store.port,
            lighthouse_addr=lh.address(),
            replica_id="v0",
            hostname="127.0.0.1",
        )
        try:
            with pytest.raises(ValueError, match="synchronous quorum"):
                DiLoCo(
                    manager,
                    [model],
                    torch.optim.SGD(model.parameters(), lr=0.1),
                    torch.optim.SGD(model.parameters(), lr=0.1),
                    sync_every=2,
                    pin_memory=False,
                )
        finally:
            manager.shutdown(wait=False)
            lh.shutdown()

    def test_recovery_after_failure(self):
        # min_replicas=2: the final window must be a JOINT commit — when
        # the faster replica exits, its heartbeat stops and the laggard
        # could otherwise commit the last outer step alone with its own
        # (replica-specific) data, a divergence no later heal repairs.
        # The survivor simply waits out the restart (~2 s) at its sync.
        lh = LighthouseServer(bind="127.0.0.1:0", min_replicas=2, join_timeout_ms=1000)
        try:
            with ThreadPoolExecutor(max_workers=2) as ex:
                futs = [
                    ex.submit(_diloco_replica, 0, lh.address(), 3, 2),
                    ex.submit(_diloco_replica, 1, lh.address(), 3, 2, 1),
                ]
                results = [f.result(timeout=120) for f in futs]
            assert results[1]["failed"]
            a, b = results[0]["original"], results[1]["original"]
            for k in a:
                torch.testing.assert_close(a[k], b[k], msg=f"mismatch at {k}")
        finally:
            lh.shutdown()


class TestSplitFragments:
    def test_balanced_split(self):
        from torchft_amd.local_sgd import split_into_fragments

        model = nn.Sequential(
            nn.Linear(8, 8), nn.Linear(8, 8), nn.Linear(8, 8), nn.Linear(8, 8)
        )
        frags = split_into_fragments(model, 2)
        assert len(frags) == 2
        total = sum(p.numel() for p in model.parameters())
        assert sum(p.numel() for f in frags for p in f.parameters()) == total
        # fragments alias the original parameters (training flows through)
        orig_ids = {id(p) for p in model.parameters()}
        frag_ids = {id(p) for f in frags for p in f.parameters()}
        assert orig_ids == frag_ids

    def test_too_many_fragments_raises(self):
        from torchft_amd.local_sgd import split_into_fragments

        with pytest.raises(ValueError):
            split_into_fragments(nn.Sequential(nn.Linear(2, 2)), 3)


class TestBucketPlan:
    """Multi-bucket packing: each bucket's scatter-back must see its own
    ranges (the reference's closure bug corrupted all but the last bucket)."""

    def test_multi_bucket_scatter_back(self):
        from torchft_amd.local_sgd import _BucketPlan

        tensors = [torch.zeros(100) for _ in range(5)]
        # cap of 2 tensors per bucket -> 3 buckets
        plan = _BucketPlan(tensors, cap_bytes=200 * 4)
        assert len(plan.buckets) == 3
        for b, slots in enumerate(plan.buckets):
            flat = plan.flatten(slots)
            flat += float(b + 1)  # simulate an allreduce result per bucket
            _BucketPlan.scatter(flat, slots)
        torch.testing.assert_close(tensors[0], torch.full((100,), 1.0))
        torch.testing.assert_close(tensors[2], torch.full((100,), 2.0))
        torch.testing.assert_close(tensors[4], torch.full((100,), 3.0))

    def test_oversized_tensor_rejected(self):
        from torchft_amd.local_sgd import _BucketPlan

        with pytest.raises(ValueError, match="bucket cap"):
            _BucketPlan([torch.zeros(100)], cap_bytes=4)


class TestSplitContainers:
    def test_modulelist_flattened(self):
        """A model whose blocks live in one ModuleList (like Llama.layers)
        must split at block granularity, not count the list as one unit."""
        from torchft_amd.local_sgd import split_into_fragments

        class Net(nn.Module):
            def __init__(self):
                super().__init__()
                self.emb = nn.Embedding(10, 8)
                self.layers = nn.ModuleList(nn.Linear(8, 8) for _ in range(6))
                self.head = nn.Linear(8, 10)

        model = Net()
        frags = split_into_fragments(model, 4)
        assert len(frags) == 4
        total = sum(p.numel() for p in model.parameters())
        assert sum(p.numel() for f in frags for p in f.parameters()) == total


class TestSyncSchedule:
    """The staggered stage/commit schedule is a pure function of
    (local_step, window, delay) — every replica computes the same phases."""

    def test_no_delay_stage_and_commit_same_step(self):
        from torchft_amd.local_sgd import _SyncSchedule

        s = _SyncSchedule(window=4, delay=0)
        stages = [i for i in range(1, 9) if s.stages_now(i)]
        commits = [i for i in range(1, 9) if s.commits_now(i)]
        assert stages == [4]
        assert commits == [4]

    def test_delay_staggers_stage_before_commit(self):
        from torchft_amd.local_sgd import _SyncSchedule

        s = _SyncSchedule(window=6, delay=2)
        stages = [i for i in range(1, 13) if s.stages_now(i)]
        commits = [i for i in range(1, 13) if s.commits_now(i)]
        # stage fires `delay` steps before the commit so the allreduce
        # overlaps `delay` inner steps
        assert stages == [4]
        assert commits == [6]
        assert commits[0] - stages[0] == 2

    def test_frozen_pure_value(self):
        import dataclasses

        from torchft_amd.local_sgd import _SyncSchedule

        s = _SyncSchedule(window=4, delay=1)
        with pytest.raises(dataclasses.FrozenInstanceError):
            s.window = 5  # type: ignore[misc]


class TestSplitFragmentsProperties:
    def test_partition_invariants(self):
        """split_into_fragments partitions the model: every parameter in
        exactly one fragment, fragment count as requested, no empties."""
        from hypothesis import given, settings
        from hypothesis import strategies as st

        from torchft_amd.local_sgd import split_into_fragments

        @settings(max_examples=30, deadline=None)
        @given(
            n_layers=st.integers(2, 12),
            n_fragments=st.integers(1, 6),
            width=st.integers(1, 8),
        )
        def run(n_layers, n_fragments, width):
            if n_fragments > n_layers:
                return
            model = nn.Sequential(
                *[nn.Linear(width, width) for _ in range(n_layers)]
            )
            frags = split_into_fragments(model, n_fragments)
            assert len(frags) == n_fragments
            seen = set()
            for f in frags:
                params = list(f.parameters())
                assert params, "empty fragment"
                for p in params:
                    assert id(p) not in seen, "parameter in two fragments"
                    seen.add(id(p))
            assert len(seen) == len(list(model.parameters()))

        run()
