"""Host-side decode/sampling utilities of the product path.

Mirror of /root/reference/cosmos_curate/pipelines/video/utils/
decoder_utils.py, hot-path subset:
- ``FrameExtractionPolicy`` / ``FrameExtractionSignature`` (:86-117) —
  identical string format (``"{policy!s}-{int(fps*1000)}"``), since these
  strings key ``clip.extracted_frames`` across stages.
- ``find_closest_indices`` (:281-312) / ``sample_closest`` (:315-386) —
  the frame-index contract, product implementation (parity vs oracle
  pinned in tests/test_product_vs_oracle.py).
- ``get_video_timestamps`` (:230-278) / ``extract_video_metadata``
  (:120-197) — via the C++ demuxer (cc_demux_*), replacing PyAV/ffprobe.

Decode itself (decode_video_cpu*, :389-580) is GPU work on the rebuild and
lives in the stages (rocDecode + HIP kernels through the C ABI); there is
deliberately no CPU decoder here.
"""

from __future__ import annotations

import dataclasses
import enum

import numpy as np
import numpy.typing as npt

from cosmos_curate_amd import hotpath

DEFAULT_TRANSCODE_BITRATE_M = 4  # decoder_utils.py:43


class FrameExtractionPolicy(enum.Enum):
    """decoder_utils.py:86-96 (same members, same str() rendering)."""

    first = 0
    middle = 1
    last = 2
    sequence = 3

    def __str__(self) -> str:  # match enum str() used in signature keys
        return f"FrameExtractionPolicy.{self.name}"


@dataclasses.dataclass(frozen=True)
class FrameExtractionSignature:
    """decoder_utils.py:99-117."""

    extraction_policy: FrameExtractionPolicy
    target_fps: float

    def to_str(self) -> str:
        return f"{self.extraction_policy!s}-{int(self.target_fps * 1000)}"


@dataclasses.dataclass
class VideoMetadata:
    """decoder_utils.py:56-83 subset (probed from the container)."""

    height: int
    width: int
    fps: float
    num_frames: int
    video_codec: str
    pixel_format: str
    video_duration: float
    bit_rate_k: int
    format_name: str = "mp4"
    audio_codec: str | None = None


def find_closest_indices(
    src: npt.NDArray[np.float32], dst: npt.NDArray[np.float32]
) -> npt.NDArray[np.int32]:
    """Nearest-index map with left-tie rule (decoder_utils.py:281-312)."""
    right = np.clip(np.searchsorted(src, dst), 1, len(src) - 1)
    left = right - 1
    take_right = np.abs(dst - src[right]) < np.abs(dst - src[left])
    out = np.where(take_right, right, left)
    out = np.where(dst >= src[-1], len(src) - 1, out)
    return out.astype(np.int32)


def sample_closest(
    src: npt.NDArray[np.float32],
    sample_rate: float,
    start: float | None = None,
    stop: float | None = None,
    endpoint: bool = True,
    dedup: bool = True,
) -> tuple[npt.NDArray[np.int32], npt.NDArray[np.int32], npt.NDArray[np.float32]]:
    """Closest-index sampling with the endpoint-epsilon rule
    (decoder_utils.py:315-386)."""
    if sample_rate <= 0:
        msg = f"Sample rate must be greater than 0, got sample_rate={sample_rate}"
        raise ValueError(msg)
    interval = 1.0 / sample_rate
    lo = float(src[0]) if start is None else start
    hi = float(src[-1]) if stop is None else stop
    grid_stop = hi + interval * 0.5 if endpoint else hi
    grid = np.arange(lo, grid_stop, interval, dtype=np.float32)
    idx = find_closest_indices(src, grid)
    if not endpoint and np.isclose(grid[-1], grid_stop):
        idx = idx[:-1]
        grid = grid[:-1]
    if dedup:
        uniq, counts = np.unique(idx, return_counts=True)
        return uniq.astype(np.int32), counts.astype(np.int32), grid
    return idx, np.ones_like(idx, dtype=np.int32), grid


def get_video_timestamps(data: bytes) -> npt.NDArray[np.float32]:
    """Sorted f32 presentation timestamps (decoder_utils.py:230-278 contract),
    via the C++ MP4 demuxer."""
    with hotpath.Demuxer(bytes(data)) as d:
        return d.timestamps()


def extract_video_metadata(data: bytes) -> VideoMetadata:
    """Container probe (decoder_utils.py:120-197 semantics, demux-backed).

    num_frames follows the reference's convention int(duration * fps)
    (decoder_utils.py:177), not the sample count.
    """
    with hotpath.Demuxer(bytes(data)) as d:
        info = d.probe()
    fps = info.avg_fps
    return VideoMetadata(
        height=int(info.height),
        width=int(info.width),
        fps=fps,
        num_frames=int(info.duration_s * fps),
        video_codec="h264" if info.codec == 0 else "hevc",
        pixel_format="yuv420p",
        video_duration=info.duration_s,
        bit_rate_k=DEFAULT_TRANSCODE_BITRATE_M * 1000,
    )
