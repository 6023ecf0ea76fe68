"""Async metric collection (reference: d9d/internals/metric_collector/collector.py:10-97).

`trigger_sync` launches every metric's all-reduce on a side HIP stream so the
collectives overlap the next step's compute; `.item()` reads happen only in
`collect_results` (host sync point).
"""

from typing import Any

import torch

from ..metric.abc import Metric


class AsyncMetricCollector:
    def __init__(self, metrics: dict[str, Metric], group=None) -> None:
        self.metrics = metrics
        self.group = group
        self._stream = torch.cuda.Stream() if torch.cuda.is_available() else None
        self._synced = False

    def update(self, name: str, *args, **kwargs) -> None:
        self.metrics[name].update(*args, **kwargs)

    def trigger_sync(self) -> None:
        if self._stream is not None:
            self._stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self._stream):
                for m in self.metrics.values():
                    m.sync(self.group)
        else:
            for m in self.metrics.values():
                m.sync(self.group)
        self._synced = True

    def collect_results(self) -> dict[str, Any]:
        if not self._synced:
            self.trigger_sync()
        if self._stream is not None:
            torch.cuda.current_stream().wait_stream(self._stream)
        out = {name: m.compute() for name, m in self.metrics.items()}
        self._synced = False
        return out

    def reset(self) -> None:
        for m in self.metrics.values():
            m.reset()

    def state_dict(self) -> dict[str, Any]:
        return {name: m.state_dict() for name, m in self.metrics.items()}

    def load_state_dict(self, sd: dict[str, Any]) -> None:
        for name, m in self.metrics.items():
            if name in sd:
                m.load_state_dict(sd[name])
