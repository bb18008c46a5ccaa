"""Control-plane soak: hundreds of quorum rounds with membership churn.

Exercises the C++ lighthouse/manager through many reconfigurations —
replicas joining, leaving, rejoining — and checks invariants: quorum ids
are monotonic, every round converges, commit barriers agree.

    python scripts/soak_coord.py [--rounds 200]
"""

import argparse
import os
import random
import sys
import threading
import time
from datetime import timedelta

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from torchft_amd import _ftcore as core


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--rounds", type=int, default=200)
    p.add_argument("--replicas", type=int, default=3)
    args = p.parse_args()

    lh = core.LighthouseServer(bind="127.0.0.1:0", min_replicas=1, join_timeout_ms=50)
    rng = random.Random(7)
    mgrs = {}
    clients = {}

    gen = {i: 0 for i in range(args.replicas)}

    def make_replica(i: int):
        # unique id per incarnation, like the Python Manager's uuid suffix —
        # a restarted process is a NEW member at the lighthouse
        gen[i] += 1
        m = core.ManagerServer(
            replica_id=f"soak{i}_g{gen[i]}",
            lighthouse_addr=lh.address(),
            hostname="127.0.0.1",
            bind="127.0.0.1:0",
            store_addr=f"s{i}",
            world_size=1,
            heartbeat_interval=timedelta(milliseconds=50),
            connect_timeout=timedelta(seconds=5),
        )
        return m, core.ManagerClient(m.address(), connect_timeout=timedelta(seconds=5))

    for i in range(args.replicas):
        mgrs[i], clients[i] = make_replica(i)

    last_qid = 0
    steps = {i: 0 for i in range(args.replicas)}
    t0 = time.time()
    for rnd in range(args.rounds):
        # membership churn every ~20 rounds: bounce one replica
        if rnd % 20 == 19:
            victim = rng.randrange(args.replicas)
            mgrs[victim].shutdown()
            time.sleep(0.05)
            mgrs[victim], clients[victim] = make_replica(victim)
            steps[victim] = 0  # restarted from scratch

        alive = sorted(mgrs)
        results = {}
        errs = {}

        def quorum(i):
            try:
                results[i] = clients[i]._quorum(
                    0, steps[i], f"meta{i}", False, timedelta(seconds=10)
                )
            except Exception as e:  # noqa: BLE001
                errs[i] = e

        ts = [threading.Thread(target=quorum, args=(i,)) for i in alive]
        [t.start() for t in ts]
        [t.join(20) for t in ts]
        assert not errs, f"round {rnd}: quorum errors {errs}"
        qids = {r.quorum_id for r in results.values()}
        assert len(qids) == 1, f"round {rnd}: inconsistent quorum ids {qids}"
        qid = qids.pop()
        assert qid >= last_qid, f"round {rnd}: quorum id went backwards"
        last_qid = qid

        # commit barrier: everyone agrees
        oks = {}
        def commit(i):
            oks[i] = clients[i].should_commit(0, steps[i], True, timedelta(seconds=10))
        ts = [threading.Thread(target=commit, args=(i,)) for i in alive]
        [t.start() for t in ts]
        [t.join(20) for t in ts]
        assert all(oks.values()), f"round {rnd}: commit veto {oks}"
        for i in alive:
            steps[i] = results[i].max_step + 1

    dt = time.time() - t0
    expected_bumps = args.rounds // 20
    assert last_qid >= expected_bumps, (
        f"membership churn should bump quorum_id >= {expected_bumps}, got {last_qid}"
    )
    print(f"soak OK: {args.rounds} rounds, final quorum_id {last_qid}, "
          f"{args.rounds/dt:.1f} rounds/s ({dt:.1f}s total)")
    for m in mgrs.values():
        m.shutdown()
    lh.shutdown()


if __name__ == "__main__":
    main()
