"""TransNetV2 windowing + scene post-processing (CPU oracle).

Restates /root/reference/cosmos_curate/pipelines/video/clipping/
transnetv2_extraction_stages.py:
- ``get_batches``          (:215-235)  100-frame windows: pad 25 front/back,
                                       stride 50, replicate-edge padding
- ``predictions_to_scenes``(:264-296, ``_get_scenes``)  0/1 array -> (S,2)
- ``filter_scenes``        (:296-346, ``_get_filtered_scenes``)
- ``crop_scenes``          (:348-363)
- ``create_spans``         (:365-412)

Pure numpy; pinned by behavior tables in tests/test_transnetv2.py that
encode the reference's documented semantics (the reference's own
integration tests assert clip counts/spans on real media, which need the
real weights — unavailable offline; the *postprocess* math here is exact).
"""

from __future__ import annotations

import numpy as np
import numpy.typing as npt


def get_batches(frames: npt.NDArray[np.uint8]):
    """Yield padded 100-frame windows, 50-frame stride (ref :215-235)."""
    total = len(frames)
    rem = -total % 50
    for i in range(0, total + rem, 50):
        lo = max(i - 25, 0)
        hi = min(i + 75, total)
        batch = frames[lo:hi]
        if i < 25:
            batch = np.concatenate([np.repeat(frames[:1], 25 - i, axis=0), batch], axis=0)
        if hi > total:
            batch = np.concatenate([batch, np.repeat(frames[-1:], hi - total, axis=0)], axis=0)
        yield batch


def predictions_to_scenes(
    predictions: npt.NDArray[np.uint8], *, entire_scene_as_clip: bool
) -> npt.NDArray[np.int32]:
    """0/1 transition flags -> scene [start,end) frame pairs (ref :264-296)."""
    scenes: list[tuple[int, int]] = []
    t_prev, start = 0, 0
    t = -1
    i = 0
    for i, t in enumerate(predictions):
        if t_prev == 1 and t == 0:
            start = i
        if t_prev == 0 and t == 1 and i != 0:
            scenes.append((start, i))
        t_prev = t
    if scenes and t == 0:
        scenes.append((start, i))
    if not scenes and entire_scene_as_clip:
        scenes.append((0, len(predictions)))
    return np.array(scenes, dtype=np.int32).reshape(-1, 2)


def create_spans(start: int, end: int, max_length: int, min_length: int | None) -> list[list[int]]:
    """Stride a long scene into max_length pieces (ref :365-412)."""
    spans = []
    pos = start
    while pos < end:
        stop = min(pos + max_length, end)
        if min_length and (stop - pos) < min_length and stop == end:
            break
        spans.append([pos, stop])
        pos = stop
    return spans


def crop_scenes(scenes: npt.NDArray[np.int32], crop_length: int) -> npt.NDArray[np.int32]:
    """Trim crop_length frames from both ends; drop empty (ref :348-363)."""
    cropped = np.stack([scenes[:, 0] + crop_length, scenes[:, 1] - crop_length]).T
    return cropped[(cropped[:, 1] - cropped[:, 0]) > 0]


def filter_scenes(
    scenes: npt.NDArray[np.int32],
    min_length: int | None = None,
    max_length: int | None = None,
    max_length_mode: str = "truncate",
    crop_length: int | None = None,
) -> npt.NDArray[np.int32]:
    """max-length truncate/stride, crop, then min-length (ref :296-346)."""
    scenes = scenes.copy()
    if max_length is not None:
        if max_length_mode == "truncate":
            scenes[:, 1] = np.minimum(scenes[:, 0] + max_length, scenes[:, 1])
        elif max_length_mode == "stride":
            out: list[list[int]] = []
            for s, e in scenes:
                out.extend(create_spans(int(s), int(e), max_length, min_length))
            scenes = np.array(out, dtype=scenes.dtype).reshape(-1, 2)
        else:
            raise NotImplementedError(max_length_mode)
    if crop_length is not None:
        scenes = crop_scenes(scenes, crop_length)
    if min_length is not None:
        scenes = scenes[(scenes[:, 1] - scenes[:, 0]) >= min_length]
    return scenes
