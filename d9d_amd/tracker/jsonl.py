"""JSONL file tracker — the offline-friendly provider (the reference ships an
Aim provider with run-hash resume, d9d/tracker/provider/aim/tracker.py:19-99;
this environment has no Aim, so runs append JSON lines and resume by run id)."""

import json
import time
import uuid
from pathlib import Path
from typing import Any

from .base import BaseTracker, BaseTrackerRun


class JsonlTrackerRun(BaseTrackerRun):
    def __init__(self, path: Path, run_id: str) -> None:
        self.path = path
        self.run_id = run_id
        self._step = 0
        self._context: dict[str, Any] = {}
        self._fh = open(path, "a")

    def set_step(self, step: int) -> None:
        self._step = step

    def set_context(self, **context: Any) -> None:
        self._context = context

    def _emit(self, record: dict) -> None:
        record.update(run=self.run_id, step=self._step, ts=time.time())
        if self._context:
            record["context"] = self._context
        self._fh.write(json.dumps(record) + "\n")
        self._fh.flush()

    def scalar(self, name: str, value: float) -> None:
        self._emit({"kind": "scalar", "name": name, "value": float(value)})

    def bins(self, name: str, values) -> None:
        self._emit({"kind": "bins", "name": name, "values": [float(v) for v in values]})

    def hparams(self, params: dict[str, Any]) -> None:
        self._emit({"kind": "hparams", "params": params})

    def close(self) -> None:
        self._fh.close()


class JsonlTracker(BaseTracker):
    def __init__(self, directory: str | Path) -> None:
        self.directory = Path(directory)
        self.directory.mkdir(parents=True, exist_ok=True)
        self._run_id: str | None = None

    def new_run(self, name: str, description: str = "") -> JsonlTrackerRun:
        if self._run_id is None:
            self._run_id = f"{name}-{uuid.uuid4().hex[:8]}"
        return JsonlTrackerRun(self.directory / f"{self._run_id}.jsonl", self._run_id)

    def state_dict(self) -> dict[str, Any]:
        return {"run_id": self._run_id}

    def load_state_dict(self, state_dict: dict[str, Any]) -> None:
        self._run_id = state_dict.get("run_id")
