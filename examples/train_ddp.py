"""Fault-tolerant DDP training example.

Reference parity: /root/reference/train_ddp.py (CIFAR-10 CNN + dummy 1 GB
embedding to stretch comms, NCCL/Gloo auto-pick, PGTransport, torch
profiler chrome traces). This environment has no dataset downloads, so the
data is synthetic CIFAR-shaped tensors; everything else matches.

Run one replica group (torchrun) per group:
    TORCHFT_LIGHTHOUSE=... REPLICA_GROUP_ID=0 NUM_REPLICA_GROUPS=2 \
        torchrun --nnodes=1 --nproc-per-node=1 examples/train_ddp.py
or use the launcher:
    python -m torchft_amd.launcher --replicas 2 -- examples/train_ddp.py
"""

from __future__ import annotations

import argparse
import os
import sys
from datetime import timedelta

import torch
import torch.nn.functional as F
from torch import nn

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from torchft_amd import (
    DistributedDataParallel,
    DistributedSampler,
    Manager,
    OptimizerWrapper,
    ProcessGroupGloo,
    ProcessGroupRCCL,
)
from torchft_amd.checkpointing.pg_transport import PGTransport


class Net(nn.Module):
    """Toy CNN + a large dummy embedding to make the allreduce non-trivial."""

    def __init__(self, comm_stress_mb: int = 64) -> None:
        super().__init__()
        self.cnn = nn.Sequential(
            nn.Conv2d(3, 32, 3, padding=1), nn.ReLU(),
            nn.Conv2d(32, 64, 3, padding=1), nn.ReLU(),
            nn.AdaptiveAvgPool2d(4), nn.Flatten(),
            nn.Linear(64 * 16, 128), nn.ReLU(), nn.Linear(128, 10),
        )
        n = comm_stress_mb * 1024 * 1024 // 4
        self.dummy = nn.Parameter(torch.zeros(n))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.cnn(x) + self.dummy[0] * 0


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=1000)
    p.add_argument("--batch", type=int, default=64)
    p.add_argument("--comm-stress-mb", type=int, default=64)
    p.add_argument("--profile-dir", default=None, help="chrome trace output dir")
    args = p.parse_args()

    replica_group_id = int(os.environ.get("REPLICA_GROUP_ID", "0"))
    num_replica_groups = int(os.environ.get("NUM_REPLICA_GROUPS", "2"))
    rank = int(os.environ.get("RANK", "0"))
    world_size = int(os.environ.get("WORLD_SIZE", "1"))

    use_cuda = torch.cuda.is_available()
    device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", "0"))) if use_cuda else torch.device("cpu")
    if use_cuda:
        torch.cuda.set_device(device)

    model = Net(args.comm_stress_mb).to(device)
    pg = (
        ProcessGroupRCCL(timeout=timedelta(seconds=60))
        if use_cuda
        else ProcessGroupGloo(timeout=timedelta(seconds=60))
    )
    transport = PGTransport(
        pg, timeout=timedelta(seconds=60), device=device,
        state_dict=lambda: model.state_dict(),
    )
    manager = Manager(
        pg=pg,
        load_state_dict=model.load_state_dict,
        state_dict=model.state_dict,
        min_replica_size=1,
        replica_id=f"train_ddp_{replica_group_id}",
        checkpoint_transport=transport,
    )

    ddp = DistributedDataParallel(manager, model)
    opt = OptimizerWrapper(manager, torch.optim.AdamW(model.parameters(), lr=1e-3))

    # synthetic CIFAR-shaped data sharded the same way a real dataset would be
    data = torch.randn(4096, 3, 32, 32)
    labels = torch.randint(0, 10, (4096,))
    ds = torch.utils.data.TensorDataset(data, labels)
    sampler = DistributedSampler(
        ds,
        replica_rank=replica_group_id,
        num_replica_groups=num_replica_groups,
        group_rank=rank,
        num_replicas=world_size,
        shuffle=True,
    )
    loader = torch.utils.data.DataLoader(ds, batch_size=args.batch, sampler=sampler)

    prof = None
    if args.profile_dir:
        prof = torch.profiler.profile(
            schedule=torch.profiler.schedule(wait=2, warmup=2, active=5, repeat=1),
            on_trace_ready=torch.profiler.tensorboard_trace_handler(args.profile_dir),
            record_shapes=True,
        )
        prof.start()

    while manager.current_step() < args.steps:
        for x, y in loader:
            x, y = x.to(device), y.to(device)
            opt.zero_grad()
            loss = F.cross_entropy(ddp(x), y)
            loss.backward()
            opt.step()
            if prof is not None:
                prof.step()
            if manager.current_step() % 50 == 0 and rank == 0:
                print(
                    f"[group {replica_group_id}] step {manager.current_step()} "
                    f"loss {loss.item():.4f} participants {manager.num_participants()}",
                    flush=True,
                )
            if manager.current_step() >= args.steps:
                break

    if prof is not None:
        prof.stop()
    manager.shutdown()


if __name__ == "__main__":
    main()
