"""Main-process-only state helpers (reference: d9d/internals/state/main_process.py).

Wrap a Stateful so its state lives only on the main rank (e.g. the tracker's
run hash): other ranks checkpoint an empty dict and ignore loads.
"""

from typing import Any

import torch.distributed as dist


def _is_main() -> bool:
    return not dist.is_initialized() or dist.get_rank() == 0


class MainProcessStateful:
    def __init__(self, inner) -> None:
        self.inner = inner

    def state_dict(self) -> dict[str, Any]:
        if _is_main():
            return self.inner.state_dict()
        return {}

    def load_state_dict(self, state: dict[str, Any]) -> None:
        if _is_main() and state:
            self.inner.load_state_dict(state)
