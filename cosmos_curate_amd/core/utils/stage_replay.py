"""Stage save / replay / compare — the production A/B parity harness.

Mirror of /root/reference/cosmos_curate/core/utils/misc/stage_replay.py and
the --stage-save/--stage-replay/--stage-compare wiring
(pipeline_interface.py:167-184, splitting_pipeline.py:941-979, comparator
registry pipelines/video/utils/data_model_compare.py; SURVEY.md §4 "this is
exactly how new-kernel vs reference outputs get compared"):

- ``StageSaveConfig`` + ``wrap_stage_for_save``: transparently pickle every
  stage's INPUT task list before process_data;
- ``replay_stage``: re-run one stage from its saved inputs;
- ``compare_stage``: re-run and diff against saved golden OUTPUTS with an
  atol + pass-rate threshold over the registered comparable fields
  (numpy/torch payloads compared numerically, scalars exactly).
"""

from __future__ import annotations

import dataclasses
import pathlib
import pickle
from typing import Any, Callable

import numpy as np

from cosmos_curate_amd.core.interfaces.stage_interface import CuratorStage, PipelineTask


@dataclasses.dataclass
class StageSaveConfig:
    """Where to save per-stage inputs (+ optionally outputs)."""

    output_path: str
    stages: list[str] | None = None  # None = all stages
    save_outputs: bool = True


class _SavingStage(CuratorStage):
    """Transparent wrapper (pipeline_interface.py:167-184 shape)."""

    def __init__(self, inner: CuratorStage, config: StageSaveConfig) -> None:
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
        d = pathlib.Path(self._config.output_path) / self._inner.name()
        d.mkdir(parents=True, exist_ok=True)
        return d

    def process_data(self, tasks: list[PipelineTask]) -> list[PipelineTask] | None:
        d = self._dir()
        (d / f"input_{self._call:04d}.pkl").write_bytes(pickle.dumps(tasks))
        out = self._inner.process_data(tasks)
        if self._config.save_outputs and out is not None:
            (d / f"output_{self._call:04d}.pkl").write_bytes(pickle.dumps(out))
        self._call += 1
        return out


def wrap_stage_for_save(stage: CuratorStage, config: StageSaveConfig | None) -> CuratorStage:
    """pipeline_interface.py:259 _conditionally_wrap_stage semantics."""
    if config is None:
        return stage
    if config.stages is not None and stage.name() not in config.stages:
        return stage
    return _SavingStage(stage, config)


def load_saved(save_dir: str, stage_name: str, kind: str = "input") -> list[list[PipelineTask]]:
    d = pathlib.Path(save_dir) / stage_name
    return [
        pickle.loads(p.read_bytes()) for p in sorted(d.glob(f"{kind}_*.pkl"))
    ]


def replay_stage(stage: CuratorStage, save_dir: str) -> list[list[PipelineTask]]:
    """Re-run a stage on its saved inputs (splitting_pipeline.py:964-979)."""
    stage.stage_setup_on_node()
    stage.stage_setup()
    outs = []
    for tasks in load_saved(save_dir, stage.name(), "input"):
        outs.append(stage.process_data(tasks))
    stage.destroy()
    return outs


# ---- comparators (data_model_compare.py shape) ---------------------------

def _to_numpy(v: Any):
    try:
        import torch

        if isinstance(v, torch.Tensor):
            return v.detach().cpu().float().numpy()
    except ImportError:
        pass
    if isinstance(v, np.ndarray):
        return v
    return None


def _iter_comparables(task: Any, prefix: str = "") -> list[tuple[str, Any]]:
    """Walk the task tree collecting numeric leaves (clips' arrays/scalars)."""
    out: list[tuple[str, Any]] = []
    videos = getattr(task, "videos", None)
    if videos is None:
        return out
    for vi, video in enumerate(videos):
        for ci, clip in enumerate(video.clips):
            base = f"{prefix}videos[{vi}].clips[{ci}]"
            out.append((base + ".span", np.array(clip.span, dtype=np.float64)))
            out.append((base + ".uuid", str(clip.uuid)))
            if clip.clip_embedding is not None:
                out.append((base + ".clip_embedding", clip.clip_embedding))
            if clip.aesthetic_score is not None:
                out.append((base + ".aesthetic_score", np.float64(clip.aesthetic_score)))
            ef = clip.extracted_frames.resolve()
            if isinstance(ef, dict):
                for k, v in sorted(ef.items()):
                    arr = _to_numpy(v)
                    if arr is not None:
                        out.append((base + f".extracted_frames[{k}]", arr))
    return out


@dataclasses.dataclass
class CompareResult:
    total: int = 0
    passed: int = 0
    failures: list[str] = dataclasses.field(default_factory=list)

    @property
    def pass_rate(self) -> float:
        return self.passed / self.total if self.total else 1.0


def compare_stage(
    stage: CuratorStage,
    save_dir: str,
    atol: float = 1e-5,
    pass_rate: float = 1.0,
    key_fn: Callable | None = None,
) -> CompareResult:
    """Re-run a stage on saved inputs; diff outputs vs saved golden outputs
    (splitting_pipeline.py:941-963 --stage-compare semantics)."""
    goldens = load_saved(save_dir, stage.name(), "output")
    replays = replay_stage(stage, save_dir)
    result = CompareResult()
    for call, (gold_tasks, got_tasks) in enumerate(zip(goldens, replays)):
        for ti, (g, r) in enumerate(zip(gold_tasks, got_tasks)):
            gl = dict(_iter_comparables(g, f"call{call}.task{ti}."))
            rl = dict(_iter_comparables(r, f"call{call}.task{ti}."))
            for key in gl:
                result.total += 1
                if key not in rl:
                    result.failures.append(f"{key}: missing in replay")
                    continue
                a, b = gl[key], rl[key]
                if isinstance(a, str) or isinstance(b, str):
                    ok = a == b
                else:
                    a = np.asarray(a, dtype=np.float64)
                    b = np.asarray(b, dtype=np.float64)
                    ok = a.shape == b.shape and bool(np.allclose(a, b, atol=atol))
                if ok:
                    result.passed += 1
                else:
                    result.failures.append(f"{key}: mismatch")
    if result.pass_rate < pass_rate:
        msg = (
            f"stage-compare failed: {result.passed}/{result.total} passed "
            f"(required {pass_rate:.2%}); first failures: {result.failures[:5]}"
        )
        raise AssertionError(msg)
    return result
