"""torchft_amd: MI355X-native per-step fault-tolerant training framework.

Capability parity with meta-pytorch/torchft (layer map in SURVEY.md), rebuilt
for AMD Instinct MI355X: C++ coordination services (lighthouse + manager),
RCCL-over-xGMI reconfigurable process groups, CDNA4 HIP kernels for the
fp8-quantized collectives and fused optimizer steps, and HIP-stream-based
live healing.
"""

# The C++ coordination core is built in-tree. On a fresh checkout it does
# not exist yet — build it on first import (g++ only, a few seconds) so
# `import torchft_amd` and `python -m torchft_amd._build` both work from
# pristine state. The HIP kernel extension stays lazy (torchft_amd.ops).
try:
    from torchft_amd import _ftcore  # noqa: F401
except ImportError:  # pragma: no cover - fresh checkout
    from torchft_amd._build import build_ftcore as _build_ftcore

    _build_ftcore()
    from torchft_amd import _ftcore  # noqa: F401

from torchft_amd.data import DistributedSampler
from torchft_amd.ddp import DistributedDataParallel
from torchft_amd.manager import Manager, WorldSizeMode
from torchft_amd.optim import OptimizerWrapper
from torchft_amd.process_group import (
    ProcessGroup,
    ProcessGroupDummy,
    ProcessGroupGloo,
    ProcessGroupNCCL,
    ProcessGroupRCCL,
)

__version__ = "0.1.0"

__all__ = [
    "DistributedDataParallel",
    "DistributedSampler",
    "Manager",
    "OptimizerWrapper",
    "ProcessGroup",
    "ProcessGroupDummy",
    "ProcessGroupGloo",
    "ProcessGroupNCCL",
    "ProcessGroupRCCL",
    "WorldSizeMode",
]

from torchft_amd.baby_process_group import (  # noqa: E402
    ProcessGroupBabyGloo,
    ProcessGroupBabyRCCL,
)
from torchft_amd.local_sgd import DiLoCo, LocalSGD, split_into_fragments  # noqa: E402
from torchft_amd.process_group import (  # noqa: E402
    ErrorSwallowingProcessGroupWrapper,
    ManagedProcessGroup,
)

__all__ += [
    "DiLoCo",
    "ErrorSwallowingProcessGroupWrapper",
    "LocalSGD",
    "ManagedProcessGroup",
    "ProcessGroupBabyGloo",
    "ProcessGroupBabyRCCL",
    "split_into_fragments",
]
