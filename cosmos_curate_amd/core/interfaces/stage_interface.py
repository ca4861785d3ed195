"""Stage plugin surface.

Byte-compatible mirror of
/root/reference/cosmos_curate/core/interfaces/stage_interface.py:
- ``PipelineTask``            (:27-58)  weight/fraction/get_major_size
- ``CuratorStageResource``    (:61-66)  cpus/gpus
- ``CuratorStage``            (:69-188) resources/model/conda_env_name/
                              stage_setup_on_node/stage_setup/process_data/
                              destroy/stage_batch_size
- ``CuratorStageSpec``        (:191-214) per-stage runner knobs

Differences from the reference, by design (MI355X rebuild):
- no Ray/cosmos-xenna types: the runner protocol lives in
  runner_interface.py and is implemented by in-repo runners;
- ``env_info``/pixi plumbing collapsed: one ROCm environment.

Stage contract highlights preserved exactly:
- ``process_data(list[PipelineTask]) -> list[PipelineTask] | None`` may
  change the task count (chunk fan-out);
- ``stage_setup`` runs in the worker process, ``__init__`` in the driver;
- per-item failures are recorded into error dicts, never raised.
"""

from __future__ import annotations

import dataclasses

from cosmos_curate_amd.core.interfaces.model_interface import ModelInterface


class PipelineTask:
    """Base class for pipeline payloads (stage_interface.py:27-58)."""

    @property
    def weight(self) -> float:
        return 1.0

    @property
    def fraction(self) -> float:
        return 1.0

    def get_major_size(self) -> int:
        return 0


@dataclasses.dataclass
class CuratorStageResource:
    """Per-worker resource request (stage_interface.py:61-66)."""

    cpus: float = 1.0
    gpus: float = 0


class CuratorStage:
    """Base class for a pipeline stage (stage_interface.py:69-188)."""

    @property
    def resources(self) -> CuratorStageResource:
        return CuratorStageResource(cpus=1.0, gpus=0.0)

    @property
    def model(self) -> ModelInterface | None:
        return None

    @property
    def conda_env_name(self) -> str | None:
        if self.model is not None:
            return self.model.conda_env_name
        return None

    def stage_setup_on_node(self) -> None:
        """Once per node, before any stage_setup (stage_interface.py:98-104)."""
        return

    def stage_setup(self) -> None:
        """Runs in the worker process (stage_interface.py:106-114)."""
        if self.model is not None:
            self.model.setup()

    def process_data(self, tasks: list[PipelineTask]) -> list[PipelineTask] | None:
        """Process a batch of tasks; may change the task count."""
        return tasks

    def destroy(self) -> None:
        return

    @property
    def stage_batch_size(self) -> int:
        return 1

    def name(self) -> str:
        return type(self).__name__


@dataclasses.dataclass
class CuratorStageSpec:
    """Stage + runner knobs (stage_interface.py:191-214).

    The xenna-specific knobs keep their names and defaults so reference
    pipeline assembly code drops in; the in-repo runners honour
    num_workers_per_node (worker count) and num_run_attempts_python
    (retry count) and accept the rest as inert tuning hints.
    """

    stage: CuratorStage
    num_workers_per_node: int | None = None
    num_run_attempts_python: int = 1
    over_provision_factor: float | None = None
    worker_max_lifetime_m: int = 0
    worker_restart_interval_m: int = 0

    def name(self) -> str:
        return self.stage.name()

    def display_str(self) -> str:
        res = self.name()
        res += f" num_workers_per_node={self.num_workers_per_node}"
        res += f" cpus={self.stage.resources.cpus}"
        res += f" gpus={self.stage.resources.gpus}"
        return res
