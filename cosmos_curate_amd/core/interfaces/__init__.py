"""Plugin surface mirror of cosmos_curate.core.interfaces (the drop-in boundary)."""

from cosmos_curate_amd.core.interfaces.model_interface import ModelInterface
from cosmos_curate_amd.core.interfaces.pipeline_interface import (
    PipelineExecutionError,
    run_pipeline,
)
from cosmos_curate_amd.core.interfaces.runner_interface import (
    RunnerInterface,
    SequentialRunner,
    WorkerPoolRunner,
)
from cosmos_curate_amd.core.interfaces.stage_interface import (
    CuratorStage,
    CuratorStageResource,
    CuratorStageSpec,
    PipelineTask,
)

__all__ = [
    "CuratorStage",
    "CuratorStageResource",
    "CuratorStageSpec",
    "ModelInterface",
    "PipelineExecutionError",
    "PipelineTask",
    "RunnerInterface",
    "SequentialRunner",
    "WorkerPoolRunner",
    "run_pipeline",
]
