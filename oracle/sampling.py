"""Frame-index sampling semantics of the reference hot path (CPU oracle).

Restates /root/reference/cosmos_curate/pipelines/video/utils/decoder_utils.py:
- ``find_closest_indices``  (decoder_utils.py:281-312)
- ``sample_closest``        (decoder_utils.py:315-386)

These two functions define the *bit-exact frame-index contract* of the
per-clip decode path: which source frames get decoded,
deduplicated, and how duplicate counts are broadcast.  The HIP product path
must reproduce the returned index/count arrays exactly.

Oracle status: pinned by the reference's own KAT tables
(test_decoder_utils.py:40-201), carried in tests/golden/sampling_kats.json.
"""

from __future__ import annotations

import numpy as np
import numpy.typing as npt


def find_closest_indices(
    src: npt.NDArray[np.float32], dst: npt.NDArray[np.float32]
) -> npt.NDArray[np.int32]:
    """For each element of ``dst``, index of the nearest element of ``src``.

    Semantics (decoder_utils.py:281-312): ties go to the LEFT neighbour;
    any dst value >= src[-1] maps to len(src)-1; the candidate pair is
    (insertion_point-1, insertion_point) with the insertion point clipped
    to [1, len(src)-1].
    """
    right = np.clip(np.searchsorted(src, dst), 1, len(src) - 1)
    left = right - 1
    # strict '<' keeps equidistant samples on the left neighbour
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
    """Sample ``src`` (sorted timestamps) at ``sample_rate`` Hz.

    Semantics (decoder_utils.py:315-386):
    - sample grid = np.arange(start, stop', 1/rate) in float32, where
      stop' = stop + 0.5/rate when endpoint=True (the endpoint-epsilon rule,
      decoder_utils.py:364-371), else stop;
    - indices = find_closest_indices(src, grid);
    - when endpoint=False and the final grid element lands on stop
      (np.isclose), it is dropped;
    - dedup=True collapses repeated indices via np.unique with counts.

    Returns (indices int32, counts int32, sample_elements float32).
    """
    if sample_rate <= 0:
        msg = f"Sample rate must be greater than 0, got sample_rate={sample_rate}"
        raise ValueError(msg)

    interval = 1.0 / sample_rate
    lo = float(src[0]) if start is None else start
    hi = float(src[-1]) if stop is None else stop
    grid_stop = hi + interval * 0.5 if endpoint else hi

    grid: npt.NDArray[np.float32] = np.arange(lo, grid_stop, interval, dtype=np.float32)
    idx = find_closest_indices(src, grid)

    if not endpoint and np.isclose(grid[-1], grid_stop):
        idx = idx[:-1]
        grid = grid[:-1]

    if dedup:
        uniq, counts = np.unique(idx, return_counts=True)
        return uniq.astype(np.int32), counts.astype(np.int32), grid
    return idx, np.ones_like(idx, dtype=np.int32), grid


def broadcast_selected(
    frames: npt.NDArray, indices: npt.NDArray[np.int32], counts: npt.NDArray[np.int32]
) -> npt.NDArray:
    """Duplicate-count broadcast of selected frames.

    Semantics of the decode loop's count broadcast
    (decoder_utils.py:447-453): output holds counts[i] copies of
    frames[indices[i]], in index order.  Used to check the product path's
    gather+broadcast against the oracle on raw-frame fixtures.
    """
    total = int(counts.sum())
    out = np.empty((total, *frames.shape[1:]), dtype=frames.dtype)
    pos = 0
    for i, c in zip(indices.tolist(), counts.tolist()):
        out[pos : pos + int(c)] = frames[i]
        pos += int(c)
    return out
