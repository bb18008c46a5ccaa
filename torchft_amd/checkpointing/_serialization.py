"""Streaming state-dict serialization for checkpoint transports.

Non-seekable stream format (sockets/HTTP bodies): a pickled header with the
pytree spec + per-leaf metadata, followed by each tensor's raw bytes in
order. Tensors never round-trip through torch.save's zip container, so a 100
GB-class MI355X state dict streams at wire speed with one staging copy.

Reference parity: torchft/checkpointing/_serialization.py (which delegates to
torch.distributed._serialization._streaming_save/load) and the
_TensorMeta/_DTensorMeta scheme of torchft/checkpointing/pg_transport.py.

SECURITY WARNING: ``streaming_load`` unpickles the header and any
non-tensor leaves — unpickling attacker-controlled bytes is arbitrary code
execution. This matches the reference's threat model (torch.load with
weights_only=False over an unauthenticated HTTP checkpoint server): the
checkpoint port MUST be reachable only from the training cluster's
private network. Do not expose HTTPTransport/ParameterServer ports to
untrusted networks.
"""

from __future__ import annotations

import pickle
import struct
from dataclasses import dataclass
from typing import IO, Any, List, Tuple, Union

import torch
from torch.utils._pytree import TreeSpec, tree_flatten, tree_unflatten

try:
    from torch.distributed.tensor import DTensor

    HAS_DTENSOR = True
except ImportError:  # pragma: no cover
    HAS_DTENSOR = False


@dataclass
class _TensorMeta:
    shape: torch.Size
    dtype: torch.dtype
    nbytes: int


@dataclass
class _DTensorMeta:
    local: _TensorMeta
    spec_bytes: bytes  # pickled DTensorSpec


@dataclass
class _PickledLeaf:
    data: bytes


LeafMeta = Union[_TensorMeta, _DTensorMeta, _PickledLeaf]


def _tensor_bytes(t: torch.Tensor) -> torch.Tensor:
    """Flat uint8 CPU view of a tensor's data (contiguous copy if needed)."""
    t = t.detach()
    if not t.is_contiguous():
        t = t.contiguous()
    if t.device.type != "cpu":
        t = t.cpu()
    if t.numel() == 0:
        return torch.empty(0, dtype=torch.uint8)
    return t.view(-1).view(torch.uint8)


def _meta_for(t: torch.Tensor) -> _TensorMeta:
    return _TensorMeta(
        shape=t.shape, dtype=t.dtype, nbytes=t.numel() * t.element_size()
    )


def split_state_dict(obj: Any) -> Tuple[List[LeafMeta], List[torch.Tensor], TreeSpec]:
    """Flatten ``obj``; return (metas, tensors-to-stream, treespec).

    ``tensors`` holds only the tensor leaves (local tensors for DTensor), in
    meta order for the tensor-typed metas.
    """
    leaves, spec = tree_flatten(obj)
    metas: List[LeafMeta] = []
    tensors: List[torch.Tensor] = []
    for leaf in leaves:
        if HAS_DTENSOR and isinstance(leaf, DTensor):
            local = leaf._local_tensor
            metas.append(
                _DTensorMeta(local=_meta_for(local), spec_bytes=pickle.dumps(leaf._spec))
            )
            tensors.append(local)
        elif isinstance(leaf, torch.Tensor):
            metas.append(_meta_for(leaf))
            tensors.append(leaf)
        else:
            metas.append(_PickledLeaf(data=pickle.dumps(leaf)))
    return metas, tensors, spec


def save_plan(obj: Any) -> Tuple[bytes, List[torch.Tensor], int]:
    """(length-prefixed header bytes, tensors to stream, total stream size)."""
    metas, tensors, spec = split_state_dict(obj)
    header = pickle.dumps((spec, metas))
    prefix = struct.pack("<Q", len(header)) + header
    total = len(prefix) + sum(
        t.numel() * t.element_size() for t in tensors
    )
    return prefix, tensors, total


def streaming_save(obj: Any, f: IO[bytes]) -> None:
    prefix, tensors, _ = save_plan(obj)
    f.write(prefix)
    for t in tensors:
        raw = _tensor_bytes(t)
        if raw.numel():
            f.write(memoryview(raw.numpy()))  # zero-copy


def _read_exact(f: IO[bytes], n: int) -> bytes:
    chunks = []
    got = 0
    while got < n:
        chunk = f.read(n - got)
        if not chunk:
            raise EOFError(f"stream ended after {got}/{n} bytes")
        chunks.append(chunk)
        got += len(chunk)
    return b"".join(chunks)


# Cap readinto slices: http.client's readinto degrades badly on very large
# buffers (~94 MB/s unbounded vs ~400+ MB/s at 4-8 MB slices, measured).
_READ_CHUNK = 8 << 20


def _readinto_tensor(f: IO[bytes], meta: _TensorMeta) -> torch.Tensor:
    t = torch.empty(meta.shape, dtype=meta.dtype)
    if meta.nbytes:
        buf = t.view(-1).view(torch.uint8).numpy()
        view = memoryview(buf)
        got = 0
        has_readinto = hasattr(f, "readinto")
        while got < meta.nbytes:
            end = min(got + _READ_CHUNK, meta.nbytes)
            if has_readinto:
                k = f.readinto(view[got:end])
                if not k:
                    raise EOFError("stream ended mid-tensor")
            else:
                data = f.read(end - got)
                if not data:
                    raise EOFError("stream ended mid-tensor")
                view[got : got + len(data)] = data
                k = len(data)
            got += k
    return t


def streaming_load(f: IO[bytes]) -> Any:
    (header_len,) = struct.unpack("<Q", _read_exact(f, 8))
    spec, metas = pickle.loads(_read_exact(f, header_len))
    leaves: List[Any] = []
    for meta in metas:
        if isinstance(meta, _PickledLeaf):
            leaves.append(pickle.loads(meta.data))
        elif isinstance(meta, _DTensorMeta):
            local = _readinto_tensor(f, meta.local)
            dt_spec = pickle.loads(meta.spec_bytes)
            assert HAS_DTENSOR
            leaves.append(
                DTensor(
                    local.requires_grad_(False),
                    dt_spec,
                    requires_grad=False,
                )
            )
        else:
            leaves.append(_readinto_tensor(f, meta))
    return tree_unflatten(leaves, spec)
