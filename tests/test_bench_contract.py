"""Guard the driver's bench.py contract.

The round-end driver launches `python bench.py --gpus N --steps K --warmup W`
and parses ONE JSON line from stdout. This runs the debug model on CPU
through the full FT stack (lighthouse + manager + managed DDP + commit
barrier) and validates every field the driver reads.
"""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(300)
def test_bench_json_contract_cpu():
    proc = subprocess.run(
        [
            sys.executable, os.path.join(REPO, "bench.py"),
            "--model", "debug", "--steps", "2", "--warmup", "1",
            "--batch", "1", "--seq", "64",
        ],
        cwd=REPO,
        capture_output=True,
        text=True,
        timeout=280,
        env={**os.environ, "MASTER_PORT": "29655"},
    )
    assert proc.returncode == 0, f"bench failed:\n{proc.stderr[-2000:]}"
    json_lines = [l for l in proc.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1, f"expected exactly one JSON line:\n{proc.stdout}"
    r = json.loads(json_lines[0])

    assert r["metric"] == "goodput_tokens_per_sec"
    assert r["unit"] == "tokens/s"
    assert r["higher_is_better"] is True
    assert r["scaling"] == "weak"
    assert r["dtype"] == "bf16"
    assert r["data"] == "synthetic"
    assert r["n_gpus"] == 1 and r["steps"] == 2 and r["warmup"] == 1
    assert r["value"] > 0 and r["ms_per_step"] > 0
    # value is the whole-job aggregate: tokens/step = batch*seq*world
    expected = 1 * 64 * 1 * 2 / (r["ms_per_step"] * 2 / 1000)
    assert abs(r["value"] - expected) / expected < 1e-6
    cfg = r["config"]
    assert cfg["model"] == "debug"
    assert cfg["global_batch"] == 1 and cfg["seq_len"] == 64
    assert cfg["parallelism"] == "ft-dp1"
    assert cfg["fault_tolerance"] is True
    # the FT path really ran: a finite loss came through the managed step
    assert isinstance(cfg["loss"], float) and cfg["loss"] == cfg["loss"]
