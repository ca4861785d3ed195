"""Hello-world pipeline — the plumbing smoke (BASELINE config #1).

Mirror of /root/reference/cosmos_curate/pipelines/examples/
hello_world_pipeline.py:39-131: a 3-stage text pipeline exercising the
task/stage/runner plumbing (task dataclass, per-stage resources, a
model-backed stage with driver-side construction + worker-side setup).
The reference's GPT2 stage becomes a tiny deterministic "model" so the
example needs no weights and runs anywhere.
"""

from __future__ import annotations

import dataclasses
import os

from cosmos_curate_amd.core.interfaces import (
    CuratorStage,
    CuratorStageResource,
    CuratorStageSpec,
    ModelInterface,
    run_pipeline,
)
from cosmos_curate_amd.core.interfaces.stage_interface import PipelineTask

EXAMPLE_PROMPTS = ["The KEY TO A CREATING GOOD art is", "Once upon a time"]


@dataclasses.dataclass
class HelloWorldTask(PipelineTask):
    """hello_world_pipeline.py:39-48."""

    prompt: str
    output: str | None = None


class _LowerCaseStage(CuratorStage):
    """hello_world_pipeline.py:60-74 (cpus=1.0)."""

    @property
    def resources(self) -> CuratorStageResource:
        return CuratorStageResource(cpus=1.0, gpus=0.0)

    def process_data(self, tasks: list[HelloWorldTask]) -> list[HelloWorldTask]:
        for task in tasks:
            task.prompt = task.prompt.lower()
        return tasks


class _PrintStage(CuratorStage):
    """hello_world_pipeline.py:76-90 (cpus=0.5)."""

    @property
    def resources(self) -> CuratorStageResource:
        return CuratorStageResource(cpus=0.5, gpus=0.0)

    def process_data(self, tasks: list[HelloWorldTask]) -> list[HelloWorldTask]:
        for task in tasks:
            print(task.prompt)
        return tasks


class _EchoModel(ModelInterface):
    """Stand-in for the GPT2 model (models/gpt2.py): deterministic text fn."""

    def __init__(self) -> None:
        self._ready = False

    @property
    def conda_env_name(self) -> str:
        return "unified"

    @property
    def model_id_names(self) -> list[str]:
        return ["echo"]

    def setup(self) -> None:
        self._ready = True

    def generate(self, prompt: str) -> str:
        assert self._ready, "setup() not called (stage_setup runs in worker)"
        return f"{prompt} [pid={os.getpid()}]"


class _ModelStage(CuratorStage):
    """hello_world_pipeline.py:92-114 shape (model built in driver,
    setup in worker, gpus fraction on the spec)."""

    def __init__(self) -> None:
        self._model = _EchoModel()

    @property
    def resources(self) -> CuratorStageResource:
        return CuratorStageResource(cpus=1.0, gpus=0.0)

    @property
    def model(self) -> ModelInterface:
        return self._model

    def process_data(self, tasks: list[HelloWorldTask]) -> list[HelloWorldTask]:
        for task in tasks:
            task.output = self._model.generate(task.prompt)
        return tasks


def main() -> list[HelloWorldTask]:
    """hello_world_pipeline.py:117-131."""
    tasks = [HelloWorldTask(prompt=x) for x in EXAMPLE_PROMPTS]
    stages: list[CuratorStage | CuratorStageSpec] = [
        CuratorStageSpec(_LowerCaseStage(), num_workers_per_node=2),
        _PrintStage(),
        _ModelStage(),
    ]
    return run_pipeline(tasks, stages)


if __name__ == "__main__":
    for t in main():
        print(t.output)
