"""Low-level coordination API for custom fault-tolerance algorithms.

Re-exports the C++ coordination core (reference parity:
torchft/coordination.py:23-39).
"""

from torchft_amd._ftcore import (
    LighthouseClient,
    LighthouseServer,
    ManagerClient,
    ManagerServer,
    Quorum,
    QuorumMember,
    QuorumResult,
)

__all__ = [
    "LighthouseClient",
    "LighthouseServer",
    "ManagerClient",
    "ManagerServer",
    "Quorum",
    "QuorumMember",
    "QuorumResult",
]
