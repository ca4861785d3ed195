"""run_pipeline: normalize stages -> specs -> runner.

Mirror of /root/reference/cosmos_curate/core/interfaces/
pipeline_interface.py:281-329: wrap bare stages into CuratorStageSpec, fill
defaults, delegate to the runner (default = the distribution-capable
runner), wrap failures in PipelineExecutionError.  Weight download and
STREAMING/BATCH selection (:120-164) collapse on the rebuild: weights are
local fixed-seed tensors and the in-repo runners have a single mode.
"""

from __future__ import annotations

import argparse
from collections.abc import Sequence
from typing import TypeVar

from cosmos_curate_amd.core.interfaces.runner_interface import (
    RunnerInterface,
    SequentialRunner,
)
from cosmos_curate_amd.core.interfaces.stage_interface import (
    CuratorStage,
    CuratorStageSpec,
    PipelineTask,
)

MODEL_WEIGHTS_PREFIX = "local://model_weights"

T = TypeVar("T", bound=PipelineTask)


class PipelineExecutionError(RuntimeError):
    """Raised when pipeline execution fails (pipeline_interface.py)."""

    def __init__(self, message: str, original_error: Exception | None) -> None:
        super().__init__(message)
        self.original_error = original_error


def _build_pipeline_stage_specs(
    stages: Sequence[CuratorStage | CuratorStageSpec],
    stage_save_config=None,
) -> list[CuratorStageSpec]:
    """Normalize to specs with defaults (pipeline_interface.py:255-275);
    optionally wrap stages for --stage-save (:259, stage_replay harness)."""
    from cosmos_curate_amd.core.utils.stage_replay import wrap_stage_for_save

    specs: list[CuratorStageSpec] = []
    for stage in stages:
        stage = (
            wrap_stage_for_save(stage, stage_save_config)
            if isinstance(stage, CuratorStage)
            else stage
        )
        if isinstance(stage, CuratorStage):
            specs.append(CuratorStageSpec(stage))
        elif isinstance(stage, CuratorStageSpec):
            specs.append(stage)
        else:
            msg = f"Invalid stage type: {type(stage)}. Expected CuratorStage or CuratorStageSpec."
            raise PipelineExecutionError(msg, original_error=None)
    for spec in specs:
        # GPU stages get retry + lifetime defaults (pipeline_interface.py:187-219)
        if spec.stage.resources.gpus > 0 and spec.worker_max_lifetime_m == 0:
            spec.worker_max_lifetime_m = 120
            spec.worker_restart_interval_m = 5
    return specs


def run_pipeline(
    input_tasks: list[T],
    stages: Sequence[CuratorStage | CuratorStageSpec],
    model_weights_prefix: str = MODEL_WEIGHTS_PREFIX,
    runner: RunnerInterface | None = None,
    stage_save_config=None,
    args: argparse.Namespace | None = None,
) -> list[T]:
    """Run the pipeline (pipeline_interface.py:281-329 contract)."""
    if runner is None:
        runner = SequentialRunner()
    stage_specs = _build_pipeline_stage_specs(stages, stage_save_config)
    execution_mode = getattr(args, "execution_mode", "AUTO") if args is not None else "AUTO"
    try:
        output_tasks = runner.run(input_tasks, stage_specs, model_weights_prefix, execution_mode)
    except Exception as e:
        msg = f"Pipeline execution failed: {e!s}"
        raise PipelineExecutionError(msg, original_error=e) from e
    return [] if output_tasks is None else output_tasks
