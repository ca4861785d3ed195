"""Ingest stage — local-filesystem flavor.

Mirror of /root/reference/cosmos_curate/pipelines/video/read_write/
download_stages.py:44-260 (``VideoDownloader``: same constructor signature
and 0.25-CPU resource shape; reads bytes into ``video.encoded_data``, then
``populate_metadata`` + ``populate_timestamps``).  S3/Azure clients are out
of hot-path scope (SURVEY.md §2 "Storage/config/db utils": local-fs subset
only); ``input_s3_profile_name`` is accepted and ignored.  Raw-NV12
payloads (the null-codec backend) are probed from their own header.
"""

from __future__ import annotations

import pathlib

import numpy as np

from cosmos_curate_amd.core.interfaces.stage_interface import (
    CuratorStage,
    CuratorStageResource,
)
from cosmos_curate_amd.core.utils.lazy_data import LazyData
from cosmos_curate_amd.core.utils.performance_utils import StageTimer
from cosmos_curate_amd.pipelines.video.utils import raw_backend
from cosmos_curate_amd.pipelines.video.utils.data_model import SplitPipeTask, Video


class VideoDownloader(CuratorStage):
    """download_stages.py:44: fill encoded_data + metadata + timestamps."""

    def __init__(
        self,
        input_path: str,
        input_s3_profile_name: str = "",
        *,
        verbose: bool = False,
        log_stats: bool = False,
    ) -> None:
        self._timer = StageTimer(self)
        self._input_path = input_path
        self._input_s3_profile_name = input_s3_profile_name
        self._verbose = verbose
        self._log_stats = log_stats

    @property
    def resources(self) -> CuratorStageResource:
        return CuratorStageResource(cpus=0.25)

    def _populate_raw(self, video: Video, raw: bytes) -> None:
        n, h, w, fps = raw_backend.parse_header(raw)
        video.metadata.size = len(raw)
        video.metadata.height = h
        video.metadata.width = w
        video.metadata.framerate = fps
        video.metadata.num_frames = n
        video.metadata.duration = n / fps
        video.metadata.video_codec = "raw"
        video.metadata.pixel_format = "nv12"
        video.timestamps = raw_backend.timestamps(raw)

    def _process_video(self, video: Video) -> None:
        path = pathlib.Path(video.input_path)
        if not path.is_absolute():
            path = pathlib.Path(self._input_path) / path
        data = path.read_bytes()
        arr = np.frombuffer(data, dtype=np.uint8)
        video.encoded_data = LazyData(value=arr, nbytes=arr.nbytes)
        if raw_backend.is_raw_nv12(data):
            self._populate_raw(video, data)
        else:
            video.populate_metadata()
            video.populate_timestamps()

    def process_data(self, tasks: list[SplitPipeTask]) -> list[SplitPipeTask] | None:
        for task in tasks:
            self._timer.reinit(self, task.get_major_size())
            for video in task.videos:
                with self._timer.time_process():
                    try:
                        self._process_video(video)
                    except Exception as e:
                        video.errors[type(self).__name__] = str(e) or type(e).__name__
            if self._log_stats:
                name, stats = self._timer.log_stats()
                task.stage_perf[name] = stats
        return tasks
