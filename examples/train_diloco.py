"""Streaming DiLoCo training example.

Reference parity: /root/reference/train_diloco.py (MLP split into fragments,
Streaming DiLoCo with sync_every=20 / fragment_sync_delay=5). Synthetic data
(no dataset downloads in this environment).

    python -m torchft_amd.launcher --replicas 2 -- examples/train_diloco.py
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

from torchft_amd import Manager, ProcessGroupGloo, ProcessGroupRCCL
from torchft_amd.local_sgd import DiLoCo


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--outer-steps", type=int, default=50)
    p.add_argument("--sync-every", type=int, default=20)
    p.add_argument("--fragment-sync-delay", type=int, default=5)
    p.add_argument("--fragments", type=int, default=2)
    p.add_argument("--quantize", action="store_true",
                   help="fp8-quantized outer allreduce (CDNA4 kernels)")
    args = p.parse_args()

    replica_group_id = int(os.environ.get("REPLICA_GROUP_ID", "0"))
    use_cuda = torch.cuda.is_available()
    device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", "0"))) if use_cuda else torch.device("cpu")
    if use_cuda:
        torch.cuda.set_device(device)

    torch.manual_seed(42)
    layers = []
    for _ in range(args.fragments):
        layers += [nn.Linear(128, 128), nn.ReLU()]
    model = nn.Sequential(*layers, nn.Linear(128, 10)).to(device)
    # fragments = consecutive slices of the module list (the reference uses
    # torch.distributed.pipelining to split; a plain slice works identically
    # for a sequential model)
    frag_modules = [
        nn.Sequential(*list(model.children())[i * 2 : (i + 1) * 2])
        for i in range(args.fragments)
    ]
    frag_modules[-1] = nn.Sequential(*list(frag_modules[-1].children()), list(model.children())[-1])

    pg = (
        ProcessGroupRCCL(timeout=timedelta(seconds=60))
        if use_cuda
        else ProcessGroupGloo(timeout=timedelta(seconds=60))
    )
    manager = Manager(
        pg=pg,
        load_state_dict=model.load_state_dict,
        state_dict=model.state_dict,
        min_replica_size=1,
        use_async_quorum=False,  # DiLoCo requires the sync quorum path
        replica_id=f"train_diloco_{replica_group_id}",
    )

    inner_opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    outer_opts = [
        torch.optim.SGD(f.parameters(), lr=0.7, momentum=0.9, nesterov=True)
        for f in frag_modules
    ]

    diloco = DiLoCo(
        manager,
        frag_modules,
        inner_opt,
        outer_opts,
        sync_every=args.sync_every,
        fragment_sync_delay=args.fragment_sync_delay,
        should_quantize=args.quantize,
    )

    with diloco:
        step = 0
        while manager.current_step() < args.outer_steps:
            torch.manual_seed(step * 1000 + replica_group_id)
            x = torch.randn(32, 128, device=device)
            y = torch.randint(0, 10, (32,), device=device)
            inner_opt.zero_grad()
            loss = F.cross_entropy(model(x), y)
            loss.backward()
            inner_opt.step()
            step += 1
            if step % 20 == 0:
                print(
                    f"[group {replica_group_id}] inner step {step} outer "
                    f"{manager.current_step()} loss {loss.item():.4f}",
                    flush=True,
                )
    manager.shutdown()


if __name__ == "__main__":
    main()
