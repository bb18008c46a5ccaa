"""Smoke-run the shipped examples on CPU so they cannot rot.

Each example runs as a real subprocess (one replica group, world 1) against
a live lighthouse, for a handful of steps.
"""

import os
import socket
import subprocess
import sys

import pytest

from torchft_amd._ftcore import LighthouseServer

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _run_example(script: str, extra_args, lighthouse_addr: str) -> str:
    # the Manager connects to MASTER_ADDR:MASTER_PORT as a client — under
    # torchrun the elastic agent hosts that TCPStore; here the test does
    from torch.distributed import TCPStore

    store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
    env = {
        **os.environ,
        "TORCHFT_LIGHTHOUSE": lighthouse_addr,
        "REPLICA_GROUP_ID": "0",
        "NUM_REPLICA_GROUPS": "1",
        "RANK": "0",
        "WORLD_SIZE": "1",
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(store.port),
    }
    proc = subprocess.run(
        [sys.executable, os.path.join(REPO, "examples", script), *extra_args],
        cwd=REPO,
        capture_output=True,
        text=True,
        timeout=280,
        env=env,
    )
    assert proc.returncode == 0, (
        f"{script} failed:\n{proc.stderr[-3000:]}\n{proc.stdout[-1000:]}"
    )
    return proc.stdout + proc.stderr


@pytest.fixture()
def lighthouse():
    lh = LighthouseServer(bind="127.0.0.1:0", min_replicas=1, join_timeout_ms=500)
    yield lh
    lh.shutdown()


@pytest.mark.timeout(300)
def test_train_ddp_example(lighthouse):
    _run_example(
        "train_ddp.py",
        ["--steps", "3", "--batch", "8", "--comm-stress-mb", "1"],
        lighthouse.address(),
    )


@pytest.mark.timeout(300)
def test_train_diloco_example(lighthouse):
    _run_example(
        "train_diloco.py",
        ["--outer-steps", "4", "--sync-every", "2", "--fragments", "1",
         "--fragment-sync-delay", "0"],
        lighthouse.address(),
    )


@pytest.mark.timeout(300)
def test_launcher_two_replica_groups():
    """End-to-end: the launcher spawns 2 replica-group torchruns + a
    lighthouse; both groups join the quorum and train to completion."""
    proc = subprocess.run(
        [
            sys.executable, "-m", "torchft_amd.launcher",
            "--replicas", "2", "--min-replicas", "2",
            "--base-port", "29770",
            "--", "examples/train_ddp.py",
            "--steps", "3", "--batch", "8", "--comm-stress-mb", "1",
        ],
        cwd=REPO,
        capture_output=True,
        text=True,
        timeout=280,
    )
    assert proc.returncode == 0, (
        f"launcher failed:\n{proc.stderr[-3000:]}\n{proc.stdout[-1000:]}"
    )
    assert "lighthouse at" in proc.stdout


@pytest.mark.timeout(120)
def test_punisher_scrapes_and_kills():
    """The chaos punisher discovers replicas by scraping the dashboard's
    kill buttons and fires the kill endpoint — run it against a live
    victim manager in a subprocess."""
    import importlib.util

    from torchft_amd import _ftcore as core

    spec = importlib.util.spec_from_file_location(
        "punisher", os.path.join(REPO, "examples", "chaos", "punisher.py")
    )
    punisher = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(punisher)

    lh = core.LighthouseServer(bind="127.0.0.1:0", min_replicas=1, join_timeout_ms=100)
    p = None
    try:
        assert punisher.list_replicas(lh.address()) == []  # empty dashboard

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
            [sys.executable, "-c", child_code], stdout=subprocess.PIPE, text=True
        )
        assert p.stdout.readline().strip() == "in quorum"
        assert punisher.list_replicas(lh.address()) == ["victim"]
        punisher.kill_one(lh.address())
        assert p.wait(timeout=15) == 1
        p = None
    finally:
        if p is not None:
            p.kill()
            p.wait(timeout=10)
        lh.shutdown()
