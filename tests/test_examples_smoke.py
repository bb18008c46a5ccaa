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
