"""Property-based tests of compute_quorum_results invariants.

The recovery assignment is the subtlest piece of the protocol (the
reference unit-tests it exhaustively in Rust); here hypothesis fuzzes
membership/step/group_rank combinations and checks the structural
invariants every assignment must satisfy.
"""

from hypothesis import given, settings, strategies as st

from torchft_amd import _ftcore as core


def member(rid: str, step: int) -> core.QuorumMember:
    return core.QuorumMember(
        replica_id=rid, address=f"a_{rid}", store_address=f"s_{rid}", step=step,
        world_size=1,
    )


@st.composite
def quorums(draw):
    n = draw(st.integers(min_value=1, max_value=8))
    steps = draw(st.lists(st.integers(min_value=0, max_value=5), min_size=n, max_size=n))
    q = core.Quorum()
    q.quorum_id = draw(st.integers(min_value=1, max_value=100))
    q.participants = [member(f"r{i:02d}", s) for i, s in enumerate(steps)]
    return q


@settings(max_examples=200, deadline=None)
@given(q=quorums(), group_rank=st.integers(min_value=0, max_value=7),
       init_sync=st.booleans())
def test_assignment_invariants(q, group_rank, init_sync):
    n = len(q.participants)
    results = {
        p.replica_id: core.compute_quorum_results(p.replica_id, group_rank, q, init_sync)
        for p in q.participants
    }
    max_step = max(p.step for p in q.participants)
    by_rank = {r.replica_rank: rid for rid, r in results.items()}

    # ranks are a permutation of 0..n-1 and consistent across replicas
    assert sorted(by_rank) == list(range(n))
    for r in results.values():
        assert r.replica_world_size == n
        assert r.max_step == max_step
        assert r.quorum_id == q.quorum_id

    healers = {rid for rid, r in results.items() if r.heal}
    senders = {rid: r.recover_dst_replica_ranks for rid, r in results.items()
               if r.recover_dst_replica_ranks}

    for rid, r in results.items():
        if r.heal:
            # a healer has a source, the source is up-to-date and lists it
            src = r.recover_src_replica_rank
            assert src is not None
            src_id = by_rank[src]
            src_res = results[src_id]
            assert not src_res.heal, "recovery source must be up to date"
            assert r.replica_rank in src_res.recover_dst_replica_ranks
            assert r.recover_src_manager_address == f"a_{src_id}"
        else:
            assert r.recover_src_replica_rank is None

    # every dst listed by a sender is a healer, and each healer appears
    # exactly once across all senders
    listed = [d for dsts in senders.values() for d in dsts]
    assert len(listed) == len(set(listed))
    assert {by_rank[d] for d in listed} == healers

    # whoever is at max_step never heals; behind always heals
    for p in q.participants:
        if p.step < max_step:
            assert p.replica_id in healers
        elif not (init_sync and max_step == 0):
            assert p.replica_id not in healers

    # the primary store is an up-to-date replica and consistent across ranks
    stores = {r.store_address for r in results.values()}
    assert len(stores) == 1
    primary_id = stores.pop().removeprefix("s_")
    assert results[primary_id].max_replica_rank is not None


class TestParticipationFunction:
    """Invariants of the pure participation function the Manager uses
    (torchft_amd/manager.py::_participation_from)."""

    @given(
        replica_rank=st.integers(0, 15),
        world=st.integers(1, 16),
        max_rank=st.one_of(st.none(), st.integers(0, 15)),
        max_world=st.integers(1, 16),
        defer=st.booleans(),
        min_size=st.integers(1, 16),
    )
    def test_invariants(self, replica_rank, world, max_rank, max_world, defer,
                        min_size):
        from torchft_amd.manager import WorldSizeMode, _participation_from

        for mode in (WorldSizeMode.DYNAMIC, WorldSizeMode.FIXED_WITH_SPARES):
            rank, count = _participation_from(
                replica_rank=replica_rank,
                replica_world_size=world,
                max_replica_rank=max_rank,
                max_world_size=max_world,
                defer_healing=defer,
                world_size_mode=mode,
                min_replica_size=min_size,
            )
            # a participating rank always fits inside the numeric world
            if rank is not None:
                assert 0 <= rank < max(count, rank + 1)
            assert count >= 0
            if mode == WorldSizeMode.FIXED_WITH_SPARES:
                # the numeric world never exceeds the fixed size, and any
                # overflow rank is benched, never renumbered
                assert count <= min_size
                if rank is not None:
                    assert rank < min_size
            else:
                src = (max_rank, max_world) if defer else (replica_rank, world)
                assert (rank, count) == src
