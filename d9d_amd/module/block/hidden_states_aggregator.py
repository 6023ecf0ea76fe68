"""Per-layer hidden-state snapshotting across PP stages
(reference: d9d/module/block/hidden_states_aggregator/).

Modes: "none" (no snapshots) and "mean" (mean over the sequence per layer).
The aggregate rides the pipeline as an extra activation tensor of shape
(n_layers_snapshotted, B, H) so aux losses / probes on any stage can see
earlier layers' states.
"""

from typing import Literal

import torch
from torch import nn


class HiddenStatesAggregator(nn.Module):
    def __init__(self, mode: Literal["none", "mean"] = "none") -> None:
        super().__init__()
        self.mode = mode

    def reset_parameters(self) -> None:
        pass

    def initial(self, hidden_states: torch.Tensor) -> torch.Tensor | None:
        if self.mode == "none":
            return None
        B, _, H = hidden_states.shape
        return torch.empty(0, B, H, dtype=hidden_states.dtype, device=hidden_states.device)

    def append(
        self, aggregate: torch.Tensor | None, hidden_states: torch.Tensor
    ) -> torch.Tensor | None:
        if self.mode == "none":
            return aggregate
        snap = hidden_states.mean(dim=1, keepdim=False).unsqueeze(0)  # (1,B,H)
        if aggregate is None or aggregate.numel() == 0:
            return snap
        return torch.cat([aggregate, snap], dim=0)
