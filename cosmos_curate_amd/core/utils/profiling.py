"""Per-stage profiling wrapper.

Mirror of /root/reference/cosmos_curate/core/utils/infra/profiling.py
(``profiling_wrapper`` dynamically brackets every ``process_data`` with
the enabled backends, doc :16-52, wiring pipeline_interface.py:269-272),
reduced to the backends available on ROCm:

- "gpu": torch.profiler (CUDA/ROCm activity) traces per stage call,
  written as chrome traces under <output>/profile/<stage>/;
- "cpu": cProfile (the reference uses pyinstrument — not in this image),
  pstats dumps per stage call;
- roctx ranges are always on via the stages themselves
  (core/utils/roctx.py).
"""

from __future__ import annotations

import cProfile
import dataclasses
import pathlib

from cosmos_curate_amd.core.interfaces.stage_interface import CuratorStage, PipelineTask


@dataclasses.dataclass
class ProfilingConfig:
    """Which backends to enable + output root (profiling.py:60-120 shape)."""

    output_path: str
    profile_cpu: bool = False
    profile_gpu: bool = False


class _ProfilingStage(CuratorStage):
    """Transparent wrapper (profiling.py:124-260 shape)."""

    def __init__(self, inner: CuratorStage, config: ProfilingConfig) -> None:
        self._inner = inner
        self._config = config
        self._call = 0

    def name(self) -> str:
        return self._inner.name()

    @property
    def resources(self):
        return self._inner.resources

    @property
    def model(self):
        return self._inner.model

    def stage_setup(self) -> None:
        self._inner.stage_setup()

    def stage_setup_on_node(self) -> None:
        self._inner.stage_setup_on_node()

    def destroy(self) -> None:
        self._inner.destroy()

    def _dir(self) -> pathlib.Path:
        d = pathlib.Path(self._config.output_path) / "profile" / self._inner.name()
        d.mkdir(parents=True, exist_ok=True)
        return d

    def process_data(self, tasks: list[PipelineTask]) -> list[PipelineTask] | None:
        d = self._dir()
        call = self._call
        self._call += 1
        if self._config.profile_gpu:
            import torch
            from torch.profiler import ProfilerActivity, profile

            acts = [ProfilerActivity.CPU]
            if torch.cuda.is_available():
                acts.append(ProfilerActivity.CUDA)
            with profile(activities=acts) as prof:
                out = self._inner.process_data(tasks)
            prof.export_chrome_trace(str(d / f"gpu_{call:04d}.json"))
            return out
        if self._config.profile_cpu:
            pr = cProfile.Profile()
            pr.enable()
            try:
                out = self._inner.process_data(tasks)
            finally:
                pr.disable()
                pr.dump_stats(str(d / f"cpu_{call:04d}.pstats"))
            return out
        return self._inner.process_data(tasks)


def profiling_wrapper(stage: CuratorStage, config: ProfilingConfig | None) -> CuratorStage:
    """pipeline_interface.py:269-272 hook."""
    if config is None or not (config.profile_cpu or config.profile_gpu):
        return stage
    return _ProfilingStage(stage, config)
