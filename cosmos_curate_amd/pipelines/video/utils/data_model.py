"""Video pipeline payload model.

Mirror of /root/reference/cosmos_curate/pipelines/video/utils/data_model.py,
hot-path subset (SURVEY.md §2 "Video data model": KEEP, identical
semantics):
- ``VideoMetadata``  (:392-410)
- ``Clip``           (:194-343)  uuid/source_video/span/encoded_data/
                                 extracted_frames/clip_embedding/errors
- ``ClipStats``      (:345-390)
- ``Video``          (:413-593)  encoded_data/metadata/timestamps/clips/
                                 fraction/weight/populate_timestamps
- ``SplitPipeTask``  (:690-802)  videos + stage_perf + weight/fraction

Captioning/motion/AV/SAM fields are out of hot-path scope
and intentionally absent; the embedding field here is the generic
``clip_embedding`` produced by the CLIP/SigLIP stages (the rebuild's
embedders), parallel to intern_video_2_embedding upstream.
"""

from __future__ import annotations

import dataclasses
import pathlib
import sys
import uuid as uuid_mod
from typing import Any

import numpy as np
import numpy.typing as npt

from cosmos_curate_amd.core.interfaces.stage_interface import PipelineTask
from cosmos_curate_amd.core.utils.lazy_data import LazyData


def get_major_size(obj: object) -> int:
    """Recursive payload size in bytes (data_model.py:94-121 semantics).

    Counts numpy buffers, bytes and container members; used for perf
    accounting only.
    """
    seen: set[int] = set()
    stack = [obj]
    total = 0
    while stack:
        o = stack.pop()
        if id(o) in seen or o is None:
            continue
        seen.add(id(o))
        if isinstance(o, np.ndarray):
            total += o.nbytes
        elif isinstance(o, (bytes, bytearray)):
            total += len(o)
        elif isinstance(o, LazyData):
            total += o.nbytes
        elif isinstance(o, dict):
            stack.extend(o.values())
        elif isinstance(o, (list, tuple, set)):
            stack.extend(o)
        elif dataclasses.is_dataclass(o) and not isinstance(o, type):
            stack.extend(getattr(o, f.name) for f in dataclasses.fields(o))
        elif isinstance(o, str):
            total += sys.getsizeof(o)
    return total


@dataclasses.dataclass
class VideoMetadata:
    """data_model.py:392-410."""

    size: int | None = None
    height: int | None = None
    width: int | None = None
    framerate: float | None = None
    num_frames: int | None = None
    duration: float | None = None
    video_codec: str | None = None
    pixel_format: str | None = None
    audio_codec: str | None = None
    bit_rate_k: int | None = None
    format_name: str | None = None


@dataclasses.dataclass
class Clip:
    """data_model.py:194-343 hot-path subset."""

    uuid: uuid_mod.UUID
    source_video: str
    span: tuple[float, float]
    encoded_data: LazyData = dataclasses.field(default_factory=LazyData)
    # decoded frames keyed by FrameExtractionSignature.to_str()
    extracted_frames: LazyData = dataclasses.field(default_factory=LazyData)
    # embedding produced by the configured embedder (CLIP/SigLIP)
    clip_embedding: npt.NDArray[np.float32] | None = None
    clip_embedding_frames: LazyData = dataclasses.field(default_factory=LazyData)
    # aesthetic filter (data_model.py:241)
    aesthetic_score: float | None = None
    # motion filter (data_model.py:236-239)
    decoded_motion_data: Any = None
    motion_score_global_mean: float | None = None
    motion_score_per_patch_min_256: float | None = None
    errors: dict[str, str] = dataclasses.field(default_factory=dict)

    def __post_init__(self) -> None:
        self.encoded_data = LazyData.coerce(self.encoded_data)

    @property
    def duration(self) -> float:
        """Span length in seconds (data_model.py:311-318)."""
        return self.span[1] - self.span[0]

    def get_major_size(self) -> int:
        return get_major_size(self)


@dataclasses.dataclass
class ClipStats:
    """data_model.py:345-390: counters combined across tasks."""

    num_clips: int = 0
    num_filtered: int = 0
    num_passed: int = 0
    num_transcoded: int = 0
    num_with_embeddings: int = 0
    num_with_errors: int = 0
    # per-filter counters + duration rollups (data_model.py:345-390 subset
    # for the filters this rebuild ships)
    num_filtered_by_aesthetic: int = 0
    num_filtered_by_motion: int = 0
    total_clip_duration: float = 0.0
    max_clip_duration: float = 0.0

    def combine(self, other: "ClipStats") -> None:
        self.num_clips += other.num_clips
        self.num_filtered += other.num_filtered
        self.num_passed += other.num_passed
        self.num_transcoded += other.num_transcoded
        self.num_with_embeddings += other.num_with_embeddings
        self.num_filtered_by_aesthetic += other.num_filtered_by_aesthetic
        self.num_filtered_by_motion += other.num_filtered_by_motion
        self.total_clip_duration += other.total_clip_duration
        self.max_clip_duration = max(self.max_clip_duration, other.max_clip_duration)
        self.num_with_errors += other.num_with_errors


@dataclasses.dataclass
class Video:
    """data_model.py:413-593 hot-path subset."""

    input_video: pathlib.Path | str
    relative_path: str = ""
    encoded_data: LazyData = dataclasses.field(default_factory=LazyData)
    metadata: VideoMetadata = dataclasses.field(default_factory=VideoMetadata)
    frame_array: LazyData = dataclasses.field(default_factory=LazyData)
    timestamps: npt.NDArray[np.float32] | None = None
    clips: list[Clip] = dataclasses.field(default_factory=list)
    filtered_clips: list[Clip] = dataclasses.field(default_factory=list)
    num_total_clips: int = 0
    num_clip_chunks: int = 0
    clip_chunk_index: int = 0
    clip_stats: ClipStats = dataclasses.field(default_factory=ClipStats)
    was_remuxed: bool = False  # transcode produced new containers (:was_remuxed)
    errors: dict[str, str] = dataclasses.field(default_factory=dict)

    def __post_init__(self) -> None:
        self.encoded_data = LazyData.coerce(self.encoded_data)

    def populate_timestamps(self) -> None:
        """data_model.py:449-460: PTS from encoded_data via demux."""
        from cosmos_curate_amd.pipelines.video.utils import decoder_utils

        data = self.encoded_data.resolve()
        if data is None:
            msg = "No video data available: encoded_data is None"
            raise ValueError(msg)
        self.timestamps = decoder_utils.get_video_timestamps(bytes(data))

    def populate_metadata(self) -> None:
        """data_model.py:462-495: probe metadata from the container."""
        from cosmos_curate_amd.pipelines.video.utils import decoder_utils

        data = self.encoded_data.resolve()
        if data is None:
            msg = "No video data available: encoded_data is None"
            raise ValueError(msg)
        md = decoder_utils.extract_video_metadata(bytes(data))
        self.metadata.size = len(data) if isinstance(data, (bytes, bytearray)) else data.nbytes
        self.metadata.height = md.height
        self.metadata.width = md.width
        self.metadata.framerate = md.fps
        self.metadata.num_frames = md.num_frames
        self.metadata.duration = md.video_duration
        self.metadata.video_codec = md.video_codec
        self.metadata.pixel_format = md.pixel_format
        self.metadata.format_name = md.format_name
        self.metadata.bit_rate_k = md.bit_rate_k

    @property
    def fraction(self) -> float:
        """data_model.py:497-507."""
        if self.num_total_clips == 0:
            return 1.0
        return (len(self.clips) + len(self.filtered_clips)) / self.num_total_clips

    @property
    def weight(self) -> float:
        """Scheduler weight = duration/300s * fraction (data_model.py:509-522)."""
        if self.metadata.size is None:
            return 0.0
        assert self.metadata.duration is not None
        return (self.metadata.duration / 300) * self.fraction

    @property
    def input_path(self) -> str:
        if isinstance(self.input_video, pathlib.Path):
            return self.input_video.as_posix()
        return str(self.input_video)

    def has_metadata(self) -> bool:
        """data_model.py:536-552."""
        return all(
            [
                self.metadata.height,
                self.metadata.width,
                self.metadata.duration,
                self.metadata.framerate,
                self.metadata.num_frames,
                self.metadata.video_codec,
            ]
        )

    def get_major_size(self) -> int:
        return get_major_size(self)


@dataclasses.dataclass
class SplitPipeTask(PipelineTask):
    """data_model.py:690-802 subset.

    session_id = the video path for single-camera tasks (data_model.py:695);
    ``video=`` kwarg accepted for single-cam construction (:710-727).
    """

    session_id: str = ""
    videos: list[Video] = dataclasses.field(default_factory=list)
    stage_perf: dict[str, Any] = dataclasses.field(default_factory=dict)
    errors: dict[str, str] = dataclasses.field(default_factory=dict)
    _init_video: Video | None = dataclasses.field(default=None, repr=False)

    def __init__(
        self,
        session_id: str = "",
        videos: list[Video] | None = None,
        stage_perf: dict[str, Any] | None = None,
        errors: dict[str, str] | None = None,
        video: Video | None = None,
    ) -> None:
        if video is not None:
            if videos:
                msg = "Cannot specify both 'video' and 'videos' parameters"
                raise ValueError(msg)
            videos = [video]
        self.session_id = session_id
        self.videos = videos if videos is not None else []
        self.stage_perf = stage_perf if stage_perf is not None else {}
        self.errors = errors if errors is not None else {}
        if not self.session_id and self.videos:
            self.session_id = self.videos[0].input_path

    @property
    def video(self) -> Video:
        """Primary video (data_model.py:730-744, single-cam accessor)."""
        assert len(self.videos) >= 1
        return self.videos[0]

    @property
    def fraction(self) -> float:
        if not self.videos:
            return 1.0
        return self.videos[0].fraction

    @property
    def weight(self) -> float:
        return sum(v.weight for v in self.videos)

    def get_major_size(self) -> int:
        return get_major_size(self)


def check_clip_time_alignment(clips_per_video: list[list[Clip]]) -> list[int]:
    """Multicam: indices where same-index clips have differing spans
    (data_model.py:595-631).  Raises when clip counts differ."""
    if not clips_per_video:
        return []
    counts = [len(c) for c in clips_per_video]
    if not all(c == counts[0] for c in counts):
        msg = (
            f"Cannot check time alignment: videos have different clip counts "
            f"{counts}. All videos must have the same number of clips."
        )
        raise ValueError(msg)
    misaligned = []
    for i in range(counts[0]):
        spans = [clips[i].span for clips in clips_per_video]
        if not all(s == spans[0] for s in spans):
            misaligned.append(i)
    return misaligned


def assert_video_clip_alignment(videos: list[Video]) -> None:
    """Multicam sync validation (data_model.py:634-687): same processed
    count per camera, identical spans per index across clips AND
    filtered_clips."""
    if not videos:
        return
    processed = [len(v.clips) + len(v.filtered_clips) for v in videos]
    if not all(p == processed[0] for p in processed):
        msg = (
            f"Multi-cam videos have processed different numbers of clips: "
            f"{processed}."
        )
        raise ValueError(msg)
    for field in ("clips", "filtered_clips"):
        groups = [getattr(v, field) for v in videos]
        bad = check_clip_time_alignment(groups)
        if bad:
            spans = [getattr(v, field)[bad[0]].span for v in videos]
            msg = (
                f"Multi-cam {field} at index {bad[0]} have misaligned spans: "
                f"{spans}. Misaligned indices: {bad}"
            )
            raise ValueError(msg)


def assert_time_alignment(tasks: list["SplitPipeTask"]) -> None:
    """Per-task multicam validation over a chunked task list
    (data_model.py:862-877; consumed by chunk_tasks)."""
    for task in tasks:
        if len(task.videos) > 1:
            assert_video_clip_alignment(task.videos)
