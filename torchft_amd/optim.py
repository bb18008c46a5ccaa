"""Fault-tolerant optimizer wrapper (reference parity: torchft/optim.py).

``zero_grad()`` starts the (async) quorum so it overlaps the forward pass;
``step()`` only commits when the whole replica group agrees via
``should_commit``.
"""

from __future__ import annotations

from typing import TYPE_CHECKING, Any, Dict, List, Mapping, Optional

import torch
from torch.optim import Optimizer

if TYPE_CHECKING:
    from torchft_amd.manager import Manager


class OptimizerWrapper(Optimizer):
    def __init__(self, manager: "Manager", optim: Optimizer) -> None:
        self.optim = optim
        self.manager = manager

    def add_param_group(self, param_group: Dict[str, Any]) -> None:
        self.optim.add_param_group(param_group)

    def load_state_dict(self, state_dict: Dict[str, Any]) -> None:
        self.optim.load_state_dict(state_dict)

    def state_dict(self) -> Dict[str, Any]:
        return self.optim.state_dict()

    def zero_grad(self, set_to_none: bool = True) -> None:
        self.manager.start_quorum()
        self.optim.zero_grad(set_to_none)

    def step(self, closure: Optional[object] = None) -> None:
        assert closure is None, "optimizers that use closures are not supported"
        if self.manager.should_commit():
            self.optim.step()

    @property
    def param_groups(self) -> List[Dict[str, Any]]:
        return self.optim.param_groups

    @property
    def state(self) -> Mapping[torch.Tensor, object]:
        return self.optim.state
