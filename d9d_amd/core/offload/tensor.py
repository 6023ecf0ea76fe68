"""In-place storage swap between device and pinned host memory.

Preserves wrapper identity: the (D)Tensor object stays the same, only its
local storage moves (reference: d9d/core/offload/tensor.py:27-51). This is the
primitive behind Trainer.sleep()/wake() — an inference engine colocated on the
same MI355X gets the 288 GB of HBM back without re-building any module.
"""

import torch
from torch.distributed.tensor import DTensor


def _local(tensor: torch.Tensor) -> torch.Tensor:
    return tensor._local_tensor if isinstance(tensor, DTensor) else tensor


def offload_tensor(tensor: torch.Tensor) -> None:
    """Swap `tensor`'s storage to pinned host memory in place."""
    local = _local(tensor)
    if local.device.type == "cpu":
        return
    host = torch.empty_like(local, device="cpu", pin_memory=torch.cuda.is_available())
    host.copy_(local, non_blocking=False)
    local.data = host


def onload_tensor(tensor: torch.Tensor, device: torch.device) -> None:
    """Swap `tensor`'s storage back to `device` in place."""
    local = _local(tensor)
    if local.device == device:
        return
    local.data = local.data.to(device, non_blocking=False)
