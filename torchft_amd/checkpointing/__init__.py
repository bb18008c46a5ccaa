from torchft_amd.checkpointing._rwlock import RWLock
from torchft_amd.checkpointing.http_transport import HTTPTransport
from torchft_amd.checkpointing.transport import CheckpointTransport

__all__ = ["CheckpointTransport", "HTTPTransport", "RWLock"]
