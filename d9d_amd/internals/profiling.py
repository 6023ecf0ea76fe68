"""torch.profiler wrapper (reference: d9d/internals/profiling/profile.py:11-100).

Cyclic schedule (wait/warmup/active), per-rank chrome traces named by mesh
coordinate, tar.gz compression. Works unchanged on ROCm: traces show HIP
kernels and streams.
"""

import tarfile
from pathlib import Path

import torch
from torch.profiler import ProfilerActivity, profile, schedule


class Profiler:
    def __init__(
        self,
        directory: str | Path,
        rank_tag: str,
        wait: int = 1,
        warmup: int = 2,
        active: int = 3,
        repeat: int = 1,
        with_stack: bool = False,
        enabled: bool = True,
    ) -> None:
        self.directory = Path(directory)
        self.rank_tag = rank_tag
        self.enabled = enabled
        self._prof = None
        if not enabled:
            return
        self.directory.mkdir(parents=True, exist_ok=True)
        activities = [ProfilerActivity.CPU]
        if torch.cuda.is_available():
            activities.append(ProfilerActivity.CUDA)
        self._prof = profile(
            activities=activities,
            schedule=schedule(wait=wait, warmup=warmup, active=active, repeat=repeat),
            on_trace_ready=self._export,
            record_shapes=True,
            with_stack=with_stack,
        )

    def _export(self, prof) -> None:
        trace = self.directory / f"trace_{self.rank_tag}_{prof.step_num}.json"
        prof.export_chrome_trace(str(trace))
        with tarfile.open(trace.with_suffix(".json.tar.gz"), "w:gz") as tar:
            tar.add(trace, arcname=trace.name)
        trace.unlink()

    def open(self):
        if self._prof is not None:
            self._prof.__enter__()
        return self

    def close(self) -> None:
        if self._prof is not None:
            self._prof.__exit__(None, None, None)

    def step(self) -> None:
        if self._prof is not None:
            self._prof.step()

    def __enter__(self):
        return self.open()

    def __exit__(self, *exc) -> None:
        self.close()
