"""Per-stage performance accounting.

Mirror of /root/reference/cosmos_curate/core/utils/infra/
performance_utils.py (StagePerfStats :70-141, StageTimer :195-360,
dump/summarize :145-186): same field names and aggregation rules so the
clips/sec reporting in summary.json reads identically (SURVEY.md §2
"Perf accounting": KEEP).  OTel events and S3 writers are out of hot-path
scope and omitted.
"""

from __future__ import annotations

import dataclasses
import json
import pathlib
import time
from contextlib import contextmanager
from typing import TYPE_CHECKING

if TYPE_CHECKING:
    from cosmos_curate_amd.core.interfaces.stage_interface import CuratorStage

try:
    import psutil

    _PROC = psutil.Process()

    def _get_rss_mb() -> float:
        return _PROC.memory_info().rss / (1024 * 1024)

except Exception:  # pragma: no cover - psutil is installed in this image

    def _get_rss_mb() -> float:
        return 0.0


@dataclasses.dataclass
class StagePerfStats:
    """performance_utils.py:70-141 semantics: sum times, max RSS, span walls."""

    process_time: float = 0.0
    actor_idle_time: float = 0.0
    input_data_size_mb: float = 0.0
    rss_before_mb: float = 0.0
    rss_after_mb: float = 0.0
    rss_delta_mb: float = 0.0
    wall_start: float = 0.0
    wall_end: float = 0.0

    def __add__(self, other: "StagePerfStats") -> "StagePerfStats":
        return StagePerfStats(
            process_time=self.process_time + other.process_time,
            actor_idle_time=self.actor_idle_time + other.actor_idle_time,
            input_data_size_mb=self.input_data_size_mb + other.input_data_size_mb,
            rss_before_mb=max(self.rss_before_mb, other.rss_before_mb),
            rss_after_mb=max(self.rss_after_mb, other.rss_after_mb),
            rss_delta_mb=max(self.rss_delta_mb, other.rss_delta_mb),
            wall_start=(
                min(self.wall_start, other.wall_start)
                if self.wall_start and other.wall_start
                else self.wall_start or other.wall_start
            ),
            wall_end=max(self.wall_end, other.wall_end),
        )

    def __radd__(self, other):  # sum() support
        if other == 0:
            return self
        return self.__add__(other)

    def to_dict(self) -> dict[str, float]:
        return dataclasses.asdict(self)


def summarize_perf_stats(
    task_stats: list[dict[str, StagePerfStats]],
) -> dict[str, dict[str, float]]:
    """Aggregate per-task stage stats (performance_utils.py:145-163)."""
    all_stages: set[str] = set()
    for ts in task_stats:
        all_stages.update(ts)
    return {
        stage: sum(
            (x.get(stage, StagePerfStats()) for x in task_stats), StagePerfStats()
        ).to_dict()
        for stage in sorted(all_stages)
    }


def dump_and_write_perf_stats(
    task_stats: list[dict[str, StagePerfStats]], output_path: str | None
) -> None:
    """Write aggregated stats to <output_path>/performance_stats.json."""
    if not task_stats:
        return
    data = summarize_perf_stats(task_stats)
    if output_path is not None:
        p = pathlib.Path(output_path)
        p.mkdir(parents=True, exist_ok=True)
        (p / "performance_stats.json").write_text(json.dumps(data, indent=1))


class StageTimer:
    """Per-process_data stats tracker (performance_utils.py:195-360)."""

    def __init__(self, stage: "CuratorStage") -> None:
        self._stage_name = type(stage).__name__
        self._last_active_time = time.time()
        self._initialized = False
        self._reset()

    def _reset(self) -> None:
        self._num_gpus = 0.0
        self._num_cpus = 0.0
        self._durations_s: list[float] = []
        self._input_data_size_b = 0
        self._start = 0.0
        self._idle_time_s = 0.0
        self._rss_before_mb = 0.0

    def reinit(self, stage: "CuratorStage", stage_input_size: int = 1) -> None:
        self._reset()
        self._num_gpus = stage.resources.gpus
        self._num_cpus = stage.resources.cpus
        self._input_data_size_b = stage_input_size
        self._rss_before_mb = _get_rss_mb()
        self._start = time.time()
        if self._initialized:
            self._idle_time_s = self._start - self._last_active_time
        self._initialized = True

    @contextmanager
    def time_process(self, num_samples: int = 1):
        t0 = time.perf_counter()
        try:
            yield
        finally:
            self._durations_s.append(time.perf_counter() - t0)

    def log_stats(self) -> tuple[str, StagePerfStats]:
        end = time.time()
        self._last_active_time = end
        rss_after = _get_rss_mb()
        stats = StagePerfStats(
            process_time=sum(self._durations_s),
            actor_idle_time=self._idle_time_s,
            input_data_size_mb=self._input_data_size_b / (1024 * 1024),
            rss_before_mb=self._rss_before_mb,
            rss_after_mb=rss_after,
            rss_delta_mb=rss_after - self._rss_before_mb,
            wall_start=self._start,
            wall_end=end,
        )
        return self._stage_name, stats
