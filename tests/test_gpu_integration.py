"""GPU integration tests (single MI355X): the FT algorithms running with
device-resident models. Cross-replica comms use gloo (device→host bounce)
because two RCCL ranks cannot share one GPU; the compute path — fused
CDNA4 kernels, HIP streams, pinned staging — is the real one.
"""

from concurrent.futures import ThreadPoolExecutor
from datetime import timedelta
from typing import Dict

import pytest
import torch
import torch.nn as nn
from torch.distributed import TCPStore

from torchft_amd._ftcore import LighthouseServer
from torchft_amd.local_sgd import DiLoCo
from torchft_amd.manager import Manager
from torchft_amd.process_group import ProcessGroupGloo

pytestmark = pytest.mark.gpu


def _model(dev) -> nn.Module:
    m = nn.Sequential(nn.Linear(32, 64, bias=False), nn.Linear(64, 32, bias=False))
    with torch.no_grad():
        for i, p in enumerate(m.parameters()):
            vals = torch.arange(p.numel(), dtype=torch.float32).reshape(p.shape)
            p.copy_(vals * 1e-3 - 0.01 * (i + 1))
    return m.to(dev)


def _diloco_replica_gpu(replica_id: int, lighthouse_addr: str) -> Dict[str, torch.Tensor]:
    dev = torch.device("cuda:0")
    store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
    model = _model(dev)
    fragments = [model[0], model[1]]
    inner_opt = torch.optim.SGD(model.parameters(), lr=0.05)
    outer_opts = [torch.optim.SGD(f.parameters(), lr=0.5) for f in fragments]
    manager = Manager(
        pg=ProcessGroupGloo(timeout=timedelta(seconds=30)),
        load_state_dict=model.load_state_dict,
        state_dict=model.state_dict,
        min_replica_size=2,
        use_async_quorum=False,
        init_sync=False,
        rank=0,
        world_size=1,
        store_addr="127.0.0.1",
        store_port=store.port,
        lighthouse_addr=lighthouse_addr,
        replica_id=f"gdl_{replica_id}",
        hostname="127.0.0.1",
        timeout=timedelta(seconds=30),
    )
    try:
        diloco = DiLoCo(
            manager, fragments, inner_opt, outer_opts, sync_every=2, pin_memory=True
        )
        with diloco:
            step = 0
            while manager.current_step() < 2:
                x = (
                    torch.arange(4 * 32, device=dev, dtype=torch.float32).reshape(4, 32)
                    * 1e-3
                    * (replica_id + 1 + step)
                )
                inner_opt.zero_grad()
                model(x).square().mean().backward()
                inner_opt.step()
                step += 1
        torch.cuda.synchronize()
        return {
            f"{i}_{k}": v.detach().cpu().clone()
            for i, frag in enumerate(diloco._fragments)
            for k, v in frag.original_parameters.items()
        }
    finally:
        manager.shutdown(wait=False)


class TestDiLoCoOnGPU:
    def test_two_replicas_converge_on_device(self):
        lh = LighthouseServer(bind="127.0.0.1:0", min_replicas=2, join_timeout_ms=100)
        try:
            with ThreadPoolExecutor(max_workers=2) as ex:
                futs = [
                    ex.submit(_diloco_replica_gpu, i, lh.address()) for i in range(2)
                ]
                results = [f.result(timeout=120) for f in futs]
            a, b = results
            for k in a:
                torch.testing.assert_close(a[k], b[k], msg=f"mismatch at {k}")
        finally:
            lh.shutdown()


class TestCPOnGPU:
    def test_cp_attention_matches_full(self):
        import torch.nn.functional as F

        from torchft_amd.parallel.cp import cp_attention, shard_sequence

        dev = torch.device("cuda:0")
        torch.manual_seed(0)
        B, S, H, Hkv, D = 1, 128, 4, 2, 64
        q = torch.randn(B, S, H, D, device=dev, dtype=torch.bfloat16)
        k = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16)
        v = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16)
        ref = F.scaled_dot_product_attention(
            q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
            is_causal=True, enable_gqa=True,
        ).transpose(1, 2)

        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
        world = 2

        def worker(rank):
            pg = ProcessGroupGloo(timeout=timedelta(seconds=30))
            pg.configure(f"127.0.0.1:{store.port}/cpg", f"r{rank}", rank, world)
            out = cp_attention(
                shard_sequence(q, rank, world),
                shard_sequence(k, rank, world),
                shard_sequence(v, rank, world),
                pg, rank, world, True,
            )
            torch.cuda.synchronize()
            return out

        with ThreadPoolExecutor(max_workers=2) as ex:
            outs = list(ex.map(worker, range(2)))
        got = torch.cat(outs, dim=1)
        torch.testing.assert_close(got.float(), ref.float(), rtol=3e-2, atol=3e-2)


class TestCheckpointOnGPU:
    def test_http_heal_roundtrip_device(self):
        from torchft_amd.checkpointing import HTTPTransport

        dev = torch.device("cuda:0")
        src = HTTPTransport(timeout=timedelta(seconds=60))
        dst = HTTPTransport(timeout=timedelta(seconds=60))
        try:
            sd = {"w": torch.randn(1024, 1024, device=dev, dtype=torch.bfloat16)}
            src.send_checkpoint([1], step=5, state_dict=sd, timeout=timedelta(seconds=60))
            got = dst.recv_checkpoint(
                src_rank=0, metadata=src.metadata(), step=5, timeout=timedelta(seconds=60)
            )
            torch.testing.assert_close(got["w"], sd["w"].cpu())
        finally:
            src.shutdown()
            dst.shutdown()


class TestBabyRCCLOnGPU:
    def test_world1_allreduce_through_subprocess(self):
        """RCCL communicator in a child process with a HIP tensor shared via
        dmabuf IPC (HSA_ENABLE_IPC_MODE_LEGACY=0)."""
        from torchft_amd.baby_process_group import ProcessGroupBabyRCCL

        store = TCPStore("127.0.0.1", 0, is_master=True, wait_for_workers=False)
        pg = ProcessGroupBabyRCCL(timeout=120.0)
        pg.configure(f"127.0.0.1:{store.port}/babyrccl", "r0", 0, 1)
        try:
            from torch.distributed.distributed_c10d import AllreduceOptions, ReduceOp

            t = torch.full((1024,), 3.0, device="cuda:0")
            opts = AllreduceOptions()
            opts.reduceOp = ReduceOp.SUM
            pg.allreduce([t], opts).wait()
            torch.cuda.synchronize()
            torch.testing.assert_close(t, torch.full((1024,), 3.0, device="cuda:0"))
        finally:
            pg.shutdown()
