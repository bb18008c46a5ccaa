"""torchft_amd: MI355X-native per-step fault-tolerant training framework.

Capability parity with meta-pytorch/torchft (reference layer map in
SURVEY.md), rebuilt for AMD Instinct MI355X: C++ coordination services,
RCCL-over-xGMI reconfigurable process groups, CDNA4 HIP kernels for the
quantized collectives and fused optimizer steps.
"""

__version__ = "0.1.0"
