"""Goodput under injected failures — the BASELINE.json headline protocol
metric.

Runs N replica groups as subprocesses training a synthetic model through
the full FT stack, SIGKILLs a random replica with exponential inter-arrival
(--mtbf-secs), restarts it (it rejoins the quorum and live-heals), and
reports committed-batch goodput vs the healthy baseline.

CPU (gloo, debug model):  python scripts/chaos_goodput.py --duration 60
GPU: add --device cuda (one replica per GPU via HIP_VISIBLE_DEVICES), or
--device cuda --share-gpu to co-locate all replica groups on GPU 0 (the
cross-replica allreduce then runs over gloo: RCCL cannot place two ranks
on one device). --model llama --layers N trains an N-layer Llama-3-8B-
architecture model (full 4096 dim / GQA / 128k vocab) instead of the toy.
"""

from __future__ import annotations

import argparse
import json
import os
import random
import signal
import subprocess
import sys
import tempfile
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

WORKER_CODE = r"""
import os, sys, time
sys.path.insert(0, os.environ["TFT_REPO"])
from datetime import timedelta
import torch
import torch.nn as nn
from torch.distributed import TCPStore
from torchft_amd.manager import Manager
from torchft_amd.ddp import DistributedDataParallel
from torchft_amd.optim import OptimizerWrapper
from torchft_amd.process_group import ProcessGroupGloo, ProcessGroupRCCL

replica_id = int(os.environ["TFT_REPLICA_ID"])
status_file = os.environ["TFT_STATUS_FILE"]
use_cuda = os.environ.get("TFT_DEVICE", "cpu") == "cuda" and torch.cuda.is_available()
device = torch.device("cuda", 0) if use_cuda else torch.device("cpu")

torch.manual_seed(42)
if os.environ.get("TFT_MODEL", "toy") == "llama":
    from torchft_amd.models.llama import Llama, LlamaConfig

    cfg = LlamaConfig(n_layers=int(os.environ.get("TFT_LAYERS", "4")),
                      max_seq_len=int(os.environ.get("TFT_SEQ", "2048")))
    model = Llama(cfg, dtype=torch.bfloat16 if use_cuda else torch.float32,
                  checkpoint_activations=False).to(device)
    model.forward = model.forward_loss
    seq = int(os.environ.get("TFT_SEQ", "2048"))
    batch = int(os.environ.get("TFT_BATCH", "1"))

    def make_batch(step):
        g = torch.Generator().manual_seed(step)
        toks = torch.randint(0, cfg.vocab_size, (batch, seq + 1), generator=g)
        return toks[:, :-1].to(device), toks[:, 1:].to(device)

    def loss_of(ddp, step):
        x, y = make_batch(step)
        return ddp(x, y)
else:
    model = nn.Sequential(nn.Linear(64, 256), nn.ReLU(), nn.Linear(256, 64)).to(device)

    def loss_of(ddp, step):
        torch.manual_seed(step)
        x = torch.randn(32, 64, device=device)
        return ddp(x).square().mean()

store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
force_gloo = os.environ.get("TFT_FORCE_GLOO") == "1"
pg = ProcessGroupRCCL(timeout=timedelta(seconds=30)) if (use_cuda and not force_gloo) else \
     ProcessGroupGloo(timeout=timedelta(seconds=30))
manager = Manager(
    pg=pg,
    load_state_dict=model.load_state_dict,
    state_dict=model.state_dict,
    min_replica_size=1,
    rank=0, world_size=1,
    store_addr="127.0.0.1", store_port=store.port,
    lighthouse_addr=os.environ["TORCHFT_LIGHTHOUSE"],
    replica_id=f"chaos_{replica_id}",
    hostname="127.0.0.1",
    timeout=timedelta(seconds=30),
    quorum_timeout=timedelta(seconds=30),
)
ddp = DistributedDataParallel(manager, model)
opt = OptimizerWrapper(manager, torch.optim.SGD(model.parameters(), lr=0.01))
while True:
    opt.zero_grad()
    loss_of(ddp, manager.current_step()).backward()
    opt.step()
    with open(status_file, "w") as f:
        f.write(f"{manager.current_step()} {manager.batches_committed()} {time.time()}")
"""


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--replicas", type=int, default=2)
    p.add_argument("--duration", type=float, default=60.0)
    p.add_argument("--mtbf-secs", type=float, default=15.0,
                   help="mean time between kills (0 = no chaos, baseline run)")
    p.add_argument("--device", default="cpu", choices=["cpu", "cuda"])
    p.add_argument("--share-gpu", action="store_true",
                   help="all replica groups on GPU 0; cross-replica over gloo")
    p.add_argument("--model", default="toy", choices=["toy", "llama"])
    p.add_argument("--layers", type=int, default=4)
    p.add_argument("--seq", type=int, default=2048)
    p.add_argument("--batch", type=int, default=1)
    args = p.parse_args()

    from torchft_amd._ftcore import LighthouseServer

    lighthouse = LighthouseServer(bind="127.0.0.1:0", min_replicas=1, join_timeout_ms=500)
    tmp = tempfile.mkdtemp(prefix="chaos_goodput_")
    status = {i: os.path.join(tmp, f"replica_{i}.status") for i in range(args.replicas)}

    def spawn(i: int) -> subprocess.Popen:
        env = dict(os.environ)
        env.update({
            "TFT_REPO": REPO,
            "TFT_REPLICA_ID": str(i),
            "TFT_STATUS_FILE": status[i],
            "TFT_DEVICE": args.device,
            "TORCHFT_LIGHTHOUSE": lighthouse.address(),
            "TFT_MODEL": args.model,
            "TFT_LAYERS": str(args.layers),
            "TFT_SEQ": str(args.seq),
            "TFT_BATCH": str(args.batch),
        })
        if args.device == "cuda":
            if args.share_gpu:
                env["HIP_VISIBLE_DEVICES"] = "0"
                env["TFT_FORCE_GLOO"] = "1"
            else:
                env["HIP_VISIBLE_DEVICES"] = str(i)
        return subprocess.Popen([sys.executable, "-c", WORKER_CODE], env=env)

    procs = {i: spawn(i) for i in range(args.replicas)}
    rng = random.Random(1234)
    kills = 0
    t_start = time.time()
    next_kill = (t_start + rng.expovariate(1.0 / args.mtbf_secs)
                 if args.mtbf_secs > 0 else float("inf"))
    try:
        while time.time() - t_start < args.duration:
            time.sleep(0.25)
            # restart any dead replica (killed or crashed)
            for i, pr in procs.items():
                if pr.poll() is not None:
                    procs[i] = spawn(i)
            if time.time() >= next_kill:
                victim = rng.randrange(args.replicas)
                print(f"[chaos] killing replica {victim} "
                      f"(t={time.time()-t_start:.1f}s)", flush=True)
                procs[victim].send_signal(signal.SIGKILL)
                kills += 1
                next_kill = time.time() + rng.expovariate(1.0 / args.mtbf_secs)
    finally:
        for pr in procs.values():
            pr.kill()
        for pr in procs.values():
            try:
                pr.wait(timeout=10)
            except subprocess.TimeoutExpired:
                pass
        lighthouse.shutdown()

    elapsed = time.time() - t_start
    steps, batches = 0, 0
    for i, f in status.items():
        try:
            s, b, ts = open(f).read().split()
            steps = max(steps, int(s))
            batches = max(batches, int(b))
        except (OSError, ValueError):
            pass

    print(json.dumps({
        "metric": "chaos_goodput",
        "replicas": args.replicas,
        "duration_s": round(elapsed, 1),
        "mtbf_secs": args.mtbf_secs,
        "kills": kills,
        "committed_steps": steps,
        "committed_batches": batches,
        "steps_per_sec": round(steps / elapsed, 2),
        "batches_per_sec": round(batches / elapsed, 2),
        "model": args.model,
        "layers": args.layers if args.model == "llama" else None,
        "device": args.device,
    }), flush=True)


if __name__ == "__main__":
    main()
