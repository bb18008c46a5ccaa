"""HTTP checkpoint transport: recovering replicas fetch the live state dict
over HTTP from the healthy replica assigned as their recovery source.

Reference parity: torchft/checkpointing/http_transport.py (ThreadingHTTPServer
serving /checkpoint/{step}/{full|metadata|chunk_i}, CPU staging on a side
stream, RWLock step gating, optional N-way parallel chunk fetch).

MI355X notes: the device→host staging copy runs on a dedicated HIP stream
into pinned memory so it overlaps with whatever the compute stream is doing;
with 288 GB HBM per GPU a full-model stage is the dominant heal cost, so the
serve path streams raw tensor bytes (no zip container, no re-pickle of
storage) straight from the pinned staging buffers.
"""

from __future__ import annotations

import logging
import pickle
import socket
import threading
import urllib.request
from concurrent.futures import ThreadPoolExecutor
from datetime import timedelta
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Generic, List, Optional, TypeVar

import torch
from torch.utils._pytree import tree_flatten, tree_unflatten

from torchft_amd.checkpointing._rwlock import RWLock
from torchft_amd.checkpointing._serialization import (
    _DTensorMeta,
    _PickledLeaf,
    _readinto_tensor,
    _tensor_bytes,
    save_plan,
    split_state_dict,
    streaming_load,
)
from torchft_amd.checkpointing.transport import CheckpointTransport
from torchft_amd.utils import get_stream_context

logger = logging.getLogger(__name__)

T = TypeVar("T")


def _local_hostname() -> str:
    import os

    env = os.environ.get("TORCHFT_AMD_HOSTNAME")
    if env:
        return env
    host = socket.gethostname()
    try:
        socket.getaddrinfo(host, None)
        return host
    except socket.gaierror:
        return "127.0.0.1"


class _IPv6HTTPServer(ThreadingHTTPServer):
    address_family = socket.AF_INET6
    request_queue_size = 1024

    def server_bind(self) -> None:
        # dual-stack: accept IPv4 too
        try:
            self.socket.setsockopt(socket.IPPROTO_IPV6, socket.IPV6_V6ONLY, 0)
        except OSError:
            pass
        super().server_bind()


class HTTPTransport(CheckpointTransport[T], Generic[T]):
    def __init__(self, timeout: timedelta = timedelta(seconds=60), num_chunks: int = 0) -> None:
        self._timeout = timeout
        self._num_chunks = num_chunks
        self._lock = RWLock(timeout=timeout.total_seconds())
        self._allowed_step: Optional[int] = None
        self._staged: Optional[object] = None  # CPU-staged state dict
        self._staging_stream: Optional[torch.cuda.Stream] = (
            torch.cuda.Stream() if torch.cuda.is_available() else None
        )

        transport = self

        class Handler(BaseHTTPRequestHandler):
            protocol_version = "HTTP/1.1"

            def log_message(self, fmt: str, *args: object) -> None:
                logger.debug("http_transport: " + fmt % args)

            def do_GET(self) -> None:  # noqa: N802
                try:
                    parts = self.path.strip("/").split("/")
                    # /checkpoint/{step}/{what}
                    if len(parts) != 3 or parts[0] != "checkpoint":
                        self.send_error(404, "unknown path")
                        return
                    step = int(parts[1])
                    what = parts[2]
                    with transport._lock.r_lock():
                        if transport._allowed_step != step or transport._staged is None:
                            self.send_error(
                                400, f"checkpoint for step {step} not available"
                            )
                            return
                        self._serve(transport._staged, what)
                except (BrokenPipeError, ConnectionResetError):
                    pass
                except Exception as e:  # noqa: BLE001
                    logger.exception("http_transport handler failed")
                    try:
                        self.send_error(500, str(e))
                    except Exception:  # noqa: BLE001
                        pass

            def _serve(self, obj: object, what: str) -> None:
                import io

                if what == "full":
                    # stream tensors straight into the socket (zero copy,
                    # no whole-body buffer) with an exact content length
                    prefix, tensors, total = save_plan(obj)
                    self.send_response(200)
                    self.send_header("Content-Type", "application/octet-stream")
                    self.send_header("Content-Length", str(total))
                    self.end_headers()
                    self.wfile.write(prefix)
                    for t in tensors:
                        raw = _tensor_bytes(t)
                        if raw.numel():
                            self.wfile.write(memoryview(raw.numpy()))
                    return
                metas, tensors, spec = split_state_dict(obj)
                if what == "metadata":
                    payload = pickle.dumps((spec, metas))
                    self.send_response(200)
                    self.send_header("Content-Length", str(len(payload)))
                    self.end_headers()
                    self.wfile.write(payload)
                    return
                if what.startswith("chunk_"):
                    i = int(what[len("chunk_"):])
                    n = max(transport._num_chunks, 1)
                    pieces = [
                        _tensor_bytes(t)
                        for j, t in enumerate(tensors)
                        if j % n == i
                    ]
                    total = sum(p.numel() for p in pieces)
                    self.send_response(200)
                    self.send_header("Content-Length", str(total))
                    self.end_headers()
                    for p in pieces:
                        self.wfile.write(p.numpy().tobytes() if p.numel() else b"")
                    return
                self.send_error(404, f"unknown resource {what}")

        self._server = _IPv6HTTPServer(("::", 0), Handler)
        self._port = self._server.server_address[1]
        self._thread = threading.Thread(
            target=self._server.serve_forever, daemon=True, name="torchft_amd_ckpt_http"
        )
        self._thread.start()

    def metadata(self) -> str:
        return f"http://{_local_hostname()}:{self._port}"

    def _stage(self, state_dict: T) -> object:
        """Snapshot the state dict: device tensors copy to pinned CPU on the
        staging HIP stream; CPU tensors clone. The snapshot must not alias
        live training state — the owner keeps mutating it while recovering
        replicas fetch."""
        leaves, spec = tree_flatten(state_dict)
        out = []
        with get_stream_context(self._staging_stream):
            for leaf in leaves:
                if isinstance(leaf, torch.Tensor):
                    if leaf.device.type != "cpu":
                        cpu = torch.empty(
                            leaf.shape, dtype=leaf.dtype, pin_memory=True
                        )
                        cpu.copy_(leaf.detach().contiguous(), non_blocking=True)
                        out.append(cpu)
                    else:
                        out.append(leaf.detach().clone())
                else:
                    out.append(leaf)
        if self._staging_stream is not None:
            self._staging_stream.synchronize()
        return tree_unflatten(out, spec)

    def send_checkpoint(
        self, dst_ranks: List[int], step: int, state_dict: T, timeout: timedelta
    ) -> None:
        staged = self._stage(state_dict)
        with self._lock.w_lock():
            self._staged = staged
            self._allowed_step = step

    def disallow_checkpoint(self) -> None:
        """No-op by design: the staged checkpoint is a full snapshot (no
        aliasing of live training state), so serving it after training
        resumes is safe and removes the fetch-vs-next-step race the
        reference's live-serving transport has. The snapshot is replaced at
        the next send_checkpoint."""
        pass

    def _drop_checkpoint(self) -> None:
        with self._lock.w_lock():
            self._allowed_step = None
            self._staged = None

    def _open_with_retry(self, url: str, timeout: timedelta):
        """GET with retry on 400: the source stages its snapshot in its own
        quorum thread, which may lag the destination's fetch by a moment."""
        import time as _time
        import urllib.error

        deadline = _time.monotonic() + timeout.total_seconds()
        while True:
            try:
                return urllib.request.urlopen(url, timeout=timeout.total_seconds())
            except urllib.error.HTTPError as e:
                if e.code != 400 or _time.monotonic() > deadline - 0.2:
                    raise
                _time.sleep(0.05)

    def recv_checkpoint(
        self, src_rank: int, metadata: str, step: int, timeout: timedelta
    ) -> T:
        base = f"{metadata}/checkpoint/{step}"
        if self._num_chunks <= 0:
            with self._open_with_retry(f"{base}/full", timeout) as resp:
                return streaming_load(resp)

        # chunked parallel fetch: header first, then N ranges concurrently
        with self._open_with_retry(f"{base}/metadata", timeout) as resp:
            spec, metas = pickle.loads(resp.read())

        tensor_metas = [m for m in metas if not isinstance(m, _PickledLeaf)]
        results: List[Optional[torch.Tensor]] = [None] * len(tensor_metas)

        def fetch(i: int) -> None:
            with urllib.request.urlopen(
                f"{base}/chunk_{i}", timeout=timeout.total_seconds()
            ) as resp:
                for j, m in enumerate(tensor_metas):
                    if j % self._num_chunks != i:
                        continue
                    tm = m.local if isinstance(m, _DTensorMeta) else m
                    results[j] = _readinto_tensor(resp, tm)

        with ThreadPoolExecutor(max_workers=self._num_chunks) as ex:
            list(ex.map(fetch, range(self._num_chunks)))

        leaves = []
        ti = 0
        for m in metas:
            if isinstance(m, _PickledLeaf):
                leaves.append(pickle.loads(m.data))
                continue
            t = results[ti]
            assert t is not None
            ti += 1
            if isinstance(m, _DTensorMeta):
                from torch.distributed.tensor import DTensor

                leaves.append(DTensor(t, pickle.loads(m.spec_bytes), requires_grad=False))
            else:
                leaves.append(t)
        return tree_unflatten(leaves, spec)

    def shutdown(self, wait: bool = True) -> None:
        self._server.shutdown()
        if wait:
            self._thread.join(timeout=5)
