"""Output writer — minimal local-fs subset.

Mirror of /root/reference/cosmos_curate/pipelines/video/read_write/
metadata_writer_stage.py:66-420 (``ClipWriterStage``) and
summary_writers.py:271-340 (``write_split_summary``), reduced to the
hot-path deliverables (SURVEY.md §2 "Output writer": KEEP minimal —
clip payloads + per-clip metadata + embeddings parquet + summary.json):

    <output>/clips/<uuid>.bin              clip payload (mp4 or raw-NV12)
    <output>/metas/v0/<uuid>.json          span, source, errors
    <output>/clip_embd/chunk_<i>.parquet   id + embedding (pyarrow)
    <output>/summary.json                  counts + clips/sec + stage perf

The output tree names follow the reference's helpers (:167-223:
clips/, metas/v0/, <embedding>_embd/ parquet).
"""

from __future__ import annotations

import json
import os
import pathlib
import time

import numpy as np

from cosmos_curate_amd.core.interfaces.stage_interface import (
    CuratorStage,
    CuratorStageResource,
)
from cosmos_curate_amd.core.utils.performance_utils import StageTimer, summarize_perf_stats
from cosmos_curate_amd.pipelines.video.utils.data_model import SplitPipeTask


class ClipWriterStage(CuratorStage):
    """metadata_writer_stage.py:66: persist clips + metadata + embeddings."""

    def __init__(
        self,
        output_path: str,
        output_s3_profile_name: str = "",
        *,
        upload_clips: bool = True,
        upload_mp4: bool = True,
        embedding_algorithm: str = "clip",
        verbose: bool = False,
        log_stats: bool = False,
    ) -> None:
        self._timer = StageTimer(self)
        self._output_path = pathlib.Path(output_path)
        self._upload_clips = upload_clips and upload_mp4
        self._embedding_algorithm = embedding_algorithm
        self._verbose = verbose
        self._log_stats = log_stats
        self._chunk_counter = 0
        # chunk files are rank-namespaced: in a WorkerPoolRunner run one
        # process per GPU shares the output dir, and a bare per-instance
        # counter would overwrite other ranks' chunk_000000.parquet
        self._rank = int(os.environ.get("RANK", os.environ.get("LOCAL_RANK", "0")))

    @property
    def resources(self) -> CuratorStageResource:
        return CuratorStageResource(cpus=0.5)

    def stage_setup(self) -> None:
        (self._output_path / "clips").mkdir(parents=True, exist_ok=True)
        (self._output_path / "metas" / "v0").mkdir(parents=True, exist_ok=True)
        (self._output_path / f"{self._embedding_algorithm}_embd").mkdir(
            parents=True, exist_ok=True
        )

    def _write_embeddings_parquet(self, rows: list[tuple[str, np.ndarray]]) -> None:
        if not rows:
            return
        import pyarrow as pa
        import pyarrow.parquet as pq

        table = pa.table(
            {
                "id": [r[0] for r in rows],
                "embedding": [r[1].tolist() for r in rows],
            }
        )
        dest = (
            self._output_path
            / f"{self._embedding_algorithm}_embd"
            / f"chunk_{self._rank}_{self._chunk_counter:06d}.parquet"
        )
        pq.write_table(table, dest)
        self._chunk_counter += 1

    def process_data(self, tasks: list[SplitPipeTask]) -> list[SplitPipeTask] | None:
        for task in tasks:
            self._timer.reinit(self, task.get_major_size())
            with self._timer.time_process():
                emb_rows: list[tuple[str, np.ndarray]] = []
                multicam = len(task.videos) > 1
                for cam_idx, video in enumerate(task.videos):
                    for clip in video.clips:
                        # multicam cameras share clip uuids (aligned spans);
                        # per-camera outputs get a camera prefix so files
                        # don't collide (AV writer camera_id convention)
                        cid = (f"cam{cam_idx}_{clip.uuid}" if multicam
                               else str(clip.uuid))
                        if self._upload_clips and clip.encoded_data:
                            payload = clip.encoded_data.resolve()
                            with open(self._output_path / "clips" / f"{cid}.bin",
                                      "wb") as fh:
                                fh.write(payload)  # buffer protocol: no copy
                        meta = {
                            # "id" is the file/sample id (cam-prefixed for
                            # multicam); downstream consumers (sharding,
                            # dedup matching) key on it, not on the bare
                            # uuid, which multicam cameras share
                            "id": cid,
                            "uuid": str(clip.uuid),
                            "camera_index": cam_idx if multicam else None,
                            "source_video": clip.source_video,
                            "span": list(clip.span),
                            "duration": clip.duration,
                            "errors": clip.errors,
                            "has_embedding": clip.clip_embedding is not None,
                        }
                        (self._output_path / "metas" / "v0" / f"{cid}.json").write_text(
                            json.dumps(meta, indent=1)
                        )
                        if clip.clip_embedding is not None:
                            emb_rows.append((cid, clip.clip_embedding))
                            video.clip_stats.num_with_embeddings += 1
                        if clip.errors:
                            video.clip_stats.num_with_errors += 1
                        video.clip_stats.num_clips += 1
                        dur = float(clip.span[1] - clip.span[0])
                        video.clip_stats.total_clip_duration += dur
                        video.clip_stats.max_clip_duration = max(
                            video.clip_stats.max_clip_duration, dur)
                self._write_embeddings_parquet(emb_rows)
            if self._log_stats:
                name, stats = self._timer.log_stats()
                task.stage_perf[name] = stats
        return tasks


def write_split_summary(
    output_path: str,
    output_tasks: list[SplitPipeTask],
    embedding_algorithm: str = "clip",
    pipeline_run_time_s: float = 0.0,
) -> dict:
    """summary_writers.py:271-340 subset: counts + clips/sec + stage perf."""
    num_videos = sum(1 for t in output_tasks for v in t.videos if v.clip_chunk_index == 0)
    num_clips = sum(len(v.clips) for t in output_tasks for v in t.videos)
    num_with_embeddings = sum(
        1 for t in output_tasks for v in t.videos for c in v.clips
        if c.clip_embedding is not None
    )
    num_with_errors = sum(
        1 for t in output_tasks for v in t.videos for c in v.clips if c.errors
    )
    total_clip_duration = sum(
        c.duration for t in output_tasks for v in t.videos for c in v.clips
    )
    summary = {
        "timestamp": time.strftime("%Y-%m-%dT%H:%M:%S"),
        "embedding_algorithm": embedding_algorithm,
        "num_input_videos": num_videos,
        "num_clips": num_clips,
        "num_clips_with_embeddings": num_with_embeddings,
        "num_clips_with_errors": num_with_errors,
        "total_clip_duration_s": total_clip_duration,
        "pipeline_run_time_s": pipeline_run_time_s,
        "clips_per_second": (
            num_clips / pipeline_run_time_s if pipeline_run_time_s > 0 else None
        ),
        "stage_perf": summarize_perf_stats([t.stage_perf for t in output_tasks]),
    }
    out = pathlib.Path(output_path)
    out.mkdir(parents=True, exist_ok=True)
    (out / "summary.json").write_text(json.dumps(summary, indent=1))
    return summary
