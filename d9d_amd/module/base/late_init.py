"""Meta-device late-init protocol (reference: d9d/module/base/late_init.py:6).

Models are constructed on the meta device, parallelized (DTensor-ized),
materialized with `to_empty`, then `reset_parameters()` re-initializes
every leaf module in place.
"""

from typing import Protocol, runtime_checkable


@runtime_checkable
class ModuleLateInit(Protocol):
    def reset_parameters(self) -> None: ...
