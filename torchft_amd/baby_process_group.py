"""Subprocess-isolated process groups.

Runs the real communicator in a spawned child process so a wedged or
crashed RCCL communicator can never take down the trainer: the parent talks
to the child over pickled pipes (torch.multiprocessing reductions share CPU
tensors via shared memory and HIP tensors via dmabuf IPC — keep
``HSA_ENABLE_IPC_MODE_LEGACY=0``), and ``configure()`` simply kills and
respawns the child, which is the strongest possible abort.

Reference parity (semantics): torchft/process_group.py ProcessGroupBaby*
(:1356-2118). Design difference: the child synchronizes its stream before
acking a ``wait``, trading some overlap for a much simpler cross-process
contract (no event plumbing); the Manager issues waits from the comm hook
where this is off the critical path.
"""

from __future__ import annotations

import logging
import os
import queue
import threading
from datetime import timedelta
from typing import Dict, List, Optional, Union

import torch
import torch.multiprocessing as mp
from torch.distributed.distributed_c10d import (
    AllgatherOptions,
    AllreduceCoalescedOptions,
    AllreduceOptions,
    AllToAllOptions,
    BarrierOptions,
    BroadcastOptions,
    ReduceOp,
    ReduceScatterOptions,
    Work,
)
from torch.futures import Future

from torchft_amd.multiprocessing_util import _MonitoredPipe
from torchft_amd.process_group import ProcessGroup

logger = logging.getLogger(__name__)

_QUEUE_CLOSE = "queue_close"


def _pickle_safe_opts(opts: object) -> object:
    """c10d Options objects don't pickle; ship (type, fields) instead."""
    if opts is None or isinstance(opts, ReduceOp):
        return opts
    name = type(opts).__name__
    fields: Dict[str, object] = {}
    if hasattr(opts, "reduceOp"):
        fields["reduceOp"] = opts.reduceOp
    if hasattr(opts, "rootRank"):
        fields["rootRank"] = opts.rootRank
    if hasattr(opts, "timeout"):
        fields["timeout"] = opts.timeout
    return ("__c10d_opts__", name, fields)


def _unpickle_safe_opts(obj: object) -> object:
    if not (isinstance(obj, tuple) and len(obj) == 3 and obj[0] == "__c10d_opts__"):
        return obj
    _, name, fields = obj
    cls = {
        "AllreduceOptions": AllreduceOptions,
        "AllreduceCoalescedOptions": AllreduceCoalescedOptions,
        "AllgatherOptions": AllgatherOptions,
        "AllToAllOptions": AllToAllOptions,
        "BarrierOptions": BarrierOptions,
        "BroadcastOptions": BroadcastOptions,
        "ReduceScatterOptions": ReduceScatterOptions,
    }[name]
    opts = cls()
    for k, v in fields.items():
        try:
            setattr(opts, k, v)
        except Exception:  # noqa: BLE001 - field not settable on this type
            pass
    return opts


class _BabyWork(Work):
    def __init__(self, pg: "ProcessGroupBaby", op_id: int) -> None:
        super().__init__()
        self._pg = pg
        self._op_id = op_id

    def wait(self, timeout: Optional[timedelta] = None) -> bool:
        return self._pg._wait(self._op_id, timeout)

    def get_future(self) -> Future:
        return self._pg._get_future(self._op_id)


def _baby_worker(
    backend: str,
    store_addr: str,
    rank: int,
    world_size: int,
    req_pipe,
    fut_pipe,
    timeout_s: float,
) -> None:
    """Child: owns the real communicator, replays ops from the pipe."""
    try:
        if backend == "gloo":
            from torchft_amd.process_group import ProcessGroupGloo as PG
        elif backend == "rccl":
            from torchft_amd.process_group import ProcessGroupRCCL as PG
        else:
            raise ValueError(f"unknown baby backend {backend}")

        pg = PG(timeout=timedelta(seconds=timeout_s))
        pg.configure(store_addr, f"baby_{rank}", rank, world_size)
        req_pipe.send(("ready",))
    except Exception as e:  # noqa: BLE001
        req_pipe.send(RuntimeError(f"baby pg init failed: {e}"))
        return

    works: Dict[int, Work] = {}

    # Connection.send is not thread-safe: two futures completing at once
    # would interleave frames and corrupt the pipe. All completions funnel
    # through one queue drained by a single dedicated sender thread.
    fut_outbox: "queue.Queue[Optional[tuple]]" = queue.Queue()

    def _fut_sender() -> None:
        while True:
            msg = fut_outbox.get()
            if msg is None:
                return
            try:
                fut_pipe.send(msg)
            except (OSError, BrokenPipeError):
                return

    fut_sender_thread = threading.Thread(target=_fut_sender, daemon=True)
    fut_sender_thread.start()

    while True:
        try:
            cmd = req_pipe.recv()
        except (EOFError, OSError):
            break
        try:
            op = cmd[0]
            if op == _QUEUE_CLOSE:
                break
            elif op == "func":
                _, op_id, name, args, kwargs = cmd
                args = [_unpickle_safe_opts(a) for a in args]
                work = getattr(pg, name)(*args, **kwargs)
                works[op_id] = work
                req_pipe.send(("ran", op_id))
            elif op == "wait":
                _, op_id = cmd
                work = works.pop(op_id)
                work.wait()
                if torch.cuda.is_available():
                    # device-wide completion before acking: the parent's
                    # stream has no cross-process event to wait on
                    torch.cuda.current_stream().synchronize()
                req_pipe.send(("ok", op_id))
            elif op == "future":
                _, op_id = cmd
                work = works.pop(op_id)

                def _done(op_id: int = op_id, work: Work = work) -> None:
                    try:
                        work.wait()
                        if torch.cuda.is_available():
                            torch.cuda.current_stream().synchronize()
                        fut_outbox.put(("fut_ok", op_id))
                    except Exception as e:  # noqa: BLE001
                        fut_outbox.put(("fut_exc", op_id, str(e)))

                threading.Thread(target=_done, daemon=True).start()
                req_pipe.send(("ran_future", op_id))
            elif op == "num_active_work":
                req_pipe.send(len(works))
            else:
                req_pipe.send(RuntimeError(f"unknown command {op}"))
        except Exception as e:  # noqa: BLE001
            try:
                req_pipe.send(RuntimeError(f"baby op failed: {type(e).__name__}: {e}"))
            except Exception:  # noqa: BLE001
                break

    fut_outbox.put(None)  # stop the sender thread


class ProcessGroupBaby(ProcessGroup):
    """Process group running the real communicator in a subprocess."""

    _BACKEND: str = "gloo"

    def __init__(self, timeout: Union[timedelta, float] = 60.0) -> None:
        super().__init__(0, 1)
        if isinstance(timeout, timedelta):
            timeout = timeout.total_seconds()
        self._timeout: float = timeout
        self._world_size = -1
        self._proc: Optional[mp.Process] = None
        self._pipe: Optional[_MonitoredPipe] = None
        self._fut_pipe: Optional[_MonitoredPipe] = None
        self._futures: Dict[int, Future] = {}
        self._futures_lock = threading.Lock()
        self._fut_thread: Optional[threading.Thread] = None
        self._next_op_id = 0
        self._lock = threading.Lock()  # serializes request/response pairs

    def configure(
        self,
        store_addr: str,
        replica_id: str,
        rank: int,
        world_size: int,
        quorum_id: Optional[int] = None,
        group_rank: Optional[int] = None,
        group_world_size: Optional[int] = None,
        global_ranks: Optional[list[int]] = None,
    ) -> None:
        self.shutdown()  # kill-and-respawn IS the abort
        self._world_size = world_size

        ctx = mp.get_context("spawn")
        req_parent, req_child = ctx.Pipe()
        fut_parent, fut_child = ctx.Pipe()
        self._proc = ctx.Process(
            target=_baby_worker,
            args=(
                self._BACKEND,
                store_addr,
                rank,
                world_size,
                req_child,
                fut_child,
                self._timeout,
            ),
            daemon=True,
        )
        self._proc.start()
        self._pipe = _MonitoredPipe(req_parent)
        self._fut_pipe = _MonitoredPipe(fut_parent)
        ready = self._pipe.recv(self._timeout)
        assert ready == ("ready",), f"unexpected ready message: {ready}"

        self._fut_thread = threading.Thread(
            target=self._future_handler, daemon=True, name="baby_pg_futures"
        )
        self._fut_thread.start()

    def _future_handler(self) -> None:
        fut_pipe = self._fut_pipe
        assert fut_pipe is not None
        while True:
            try:
                msg = fut_pipe.recv(None)
            except (EOFError, OSError, TimeoutError):
                return
            kind = msg[0]
            op_id = msg[1]
            with self._futures_lock:
                fut = self._futures.pop(op_id, None)
            if fut is None:
                continue
            if kind == "fut_ok":
                fut.set_result(None)
            else:
                fut.set_exception(RuntimeError(msg[2]))

    def shutdown(self) -> None:
        if self._pipe is not None:
            try:
                self._pipe.send((_QUEUE_CLOSE,))
            except Exception:  # noqa: BLE001
                pass
            self._pipe.close()
            self._pipe = None
        if self._fut_pipe is not None:
            self._fut_pipe.close()
            self._fut_pipe = None
        if self._proc is not None:
            self._proc.join(timeout=2)
            if self._proc.is_alive():
                self._proc.kill()
                self._proc.join(timeout=5)
            self._proc = None
        with self._futures_lock:
            self._futures.clear()

    def abort(self) -> None:
        # killing the child aborts every in-flight collective
        if self._proc is not None:
            self._proc.kill()

    @staticmethod
    def _share_tensors(obj: object) -> object:
        if isinstance(obj, torch.Tensor):
            if obj.device.type == "cpu":
                obj.share_memory_()
            return obj
        if isinstance(obj, (list, tuple)):
            return type(obj)(ProcessGroupBaby._share_tensors(o) for o in obj)
        return obj

    def _run_func(self, name: str, *args: object, **kwargs: object) -> Work:
        with self._lock:
            pipe = self._pipe
            assert pipe is not None, "process group not configured"
            op_id = self._next_op_id
            self._next_op_id += 1
            shared = [
                _pickle_safe_opts(a) if not isinstance(a, (torch.Tensor, list, tuple, int, str))
                else self._share_tensors(a)
                for a in args
            ]
            if torch.cuda.is_available():
                # the child reads these tensors; make sure our stream wrote them
                torch.cuda.current_stream().synchronize()
            pipe.send(("func", op_id, name, shared, kwargs))
            resp = pipe.recv(self._timeout)
            assert resp == ("ran", op_id), f"unexpected response {resp}"
            return _BabyWork(self, op_id)

    def _wait(self, op_id: int, timeout: Optional[timedelta] = None) -> bool:
        with self._lock:
            pipe = self._pipe
            assert pipe is not None
            pipe.send(("wait", op_id))
            t = (timeout.total_seconds() if timeout else self._timeout)
            resp = pipe.recv(t)
            assert resp == ("ok", op_id), f"unexpected response {resp}"
        return True

    def _get_future(self, op_id: int) -> Future:
        fut: Future = Future()
        with self._futures_lock:
            self._futures[op_id] = fut
        with self._lock:
            pipe = self._pipe
            assert pipe is not None
            pipe.send(("future", op_id))
            resp = pipe.recv(self._timeout)
            assert resp == ("ran_future", op_id), f"unexpected response {resp}"
        return fut

    def num_active_work(self) -> int:
        with self._lock:
            pipe = self._pipe
            assert pipe is not None
            pipe.send(("num_active_work",))
            return int(pipe.recv(self._timeout))  # type: ignore[arg-type]

    # -- collective surface -------------------------------------------------

    def allreduce(self, tensors: List[torch.Tensor], opts: object) -> Work:
        return self._run_func("allreduce", tensors, opts)

    def allreduce_coalesced(self, tensors: List[torch.Tensor], opts: object) -> Work:
        return self._run_func("allreduce_coalesced", tensors, opts)

    def allgather(self, output_tensors, input_tensor, opts) -> Work:
        return self._run_func("allgather", output_tensors, input_tensor, opts)

    def allgather_into_tensor_coalesced(self, output_tensors, input_tensors, opts) -> Work:
        return self._run_func(
            "allgather_into_tensor_coalesced", output_tensors, input_tensors, opts
        )

    def alltoall_base(
        self, output_buffer, input_buffer, output_split_sizes, input_split_sizes, opts
    ) -> Work:
        return self._run_func(
            "alltoall_base", output_buffer, input_buffer, output_split_sizes,
            input_split_sizes, opts,
        )

    def barrier(self, opts: Optional[BarrierOptions] = None) -> Work:
        return self._run_func("barrier", opts)

    def broadcast(self, tensor_list, opts) -> Work:
        return self._run_func("broadcast", tensor_list, opts)

    def recv(self, tensors, src_rank: int, tag: int) -> Work:
        return self._run_func("recv", tensors, src_rank, tag)

    def reduce_scatter(self, output_tensors, input_tensors, opts) -> Work:
        return self._run_func("reduce_scatter", output_tensors, input_tensors, opts)

    def reduce_scatter_tensor_coalesced(self, output_tensors, input_tensors, opts) -> Work:
        return self._run_func(
            "reduce_scatter_tensor_coalesced", output_tensors, input_tensors, opts
        )

    def send(self, tensors, dst_rank: int, tag: int) -> Work:
        return self._run_func("send", tensors, dst_rank, tag)

    def size(self) -> int:
        return self._world_size

    def set_timeout(self, timeout: timedelta) -> None:
        self._timeout = timeout.total_seconds()


class ProcessGroupBabyGloo(ProcessGroupBaby):
    _BACKEND = "gloo"

    def getBackendName(self) -> str:
        return "torchft-baby-gloo"


class ProcessGroupBabyRCCL(ProcessGroupBaby):
    """RCCL in a subprocess: a wedged communicator dies with the child.

    Requires HSA_ENABLE_IPC_MODE_LEGACY=0 (dmabuf IPC) for cross-process
    HIP tensor sharing on this driver stack.
    """

    _BACKEND = "rccl"

    def getBackendName(self) -> str:
        return "torchft-baby-rccl"


ProcessGroupBabyNCCL = ProcessGroupBabyRCCL
