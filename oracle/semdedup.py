"""Semantic dedup arithmetic (CPU oracle).

Restates /root/reference/cosmos_curate/pipelines/video/dedup/
dedup_actor.py:
- ``pairwise_max_earlier``: the strict-upper-triangular max-cosine scan
  (dedup :315-460): rows sorted farthest-from-centroid first, embeddings
  L2-normalized, for each row j the maximum cosine to ANY earlier row i<j
  with its argmax index; row 0 scores (0.0, itself); "kept" =
  score <= 1 - eps (SemDeDup, arXiv:2303.09540 Table A7).
- ``kmeans``: spherical k-means in the shape the reference obtains from
  cuML KMeansMG (kmeans :182-313: L2-normalized embeddings, Euclidean
  assignment == cosine on unit vectors, centroid mean update).  cuML's
  exact iteration (kmeans++ init, oversampling) is third-party arithmetic
  not vendored in the reference — the restatement pins
  OUR product implementation (deterministic seeded init, fixed iteration
  count); parity with cuML itself is unpinned offline.
"""

from __future__ import annotations

import numpy as np
import numpy.typing as npt


def normalize_rows(e: npt.NDArray[np.float32]) -> npt.NDArray[np.float32]:
    n = np.linalg.norm(e, axis=1, keepdims=True)
    return (e / np.maximum(n, 1e-12)).astype(np.float32)


def pairwise_max_earlier(
    e_sorted: npt.NDArray[np.float32],
) -> tuple[npt.NDArray[np.float32], npt.NDArray[np.int32]]:
    """For each row j: (max cosine to rows i<j, argmax i).  Row 0 -> (0, 0).

    e_sorted must already be in scan order (farthest-from-centroid first)
    and is L2-normalized here (dedup_actor.py:399-407 semantics).
    """
    e = normalize_rows(e_sorted)
    m = len(e)
    maxv = np.full(m, -1.0, dtype=np.float32)
    argi = np.zeros(m, dtype=np.int32)
    if m == 0:
        return maxv, argi
    sims = np.clip(e @ e.T, -1.0, 1.0)
    for j in range(1, m):
        col = sims[:j, j]
        i = int(np.argmax(col))
        maxv[j] = col[i]
        argi[j] = i
    maxv[0] = 0.0
    argi[0] = 0
    return maxv, argi


def kept_mask(maxv: npt.NDArray[np.float32], eps: float) -> npt.NDArray[np.bool_]:
    """Row kept iff max-cosine <= 1 - eps (dedup_actor.py:334-336)."""
    return maxv <= np.float32(1.0 - eps)


def kmeans(
    e: npt.NDArray[np.float32],
    n_clusters: int,
    n_iters: int = 20,
    seed: int = 0x5EED,
) -> tuple[npt.NDArray[np.float32], npt.NDArray[np.int32]]:
    """Deterministic spherical k-means: seeded row-sample init, cosine
    assignment, mean update, centroid re-normalization.

    Returns (centroids [k,d] unit rows, labels [m]).
    """
    e = normalize_rows(e)
    m, d = e.shape
    rng = np.random.default_rng(seed)
    init = rng.choice(m, size=n_clusters, replace=False)
    cent = e[np.sort(init)].copy()
    labels = np.zeros(m, dtype=np.int32)
    for _ in range(n_iters):
        sims = e @ cent.T  # (m, k) cosine
        labels = np.argmax(sims, axis=1).astype(np.int32)
        sums = np.zeros((n_clusters, d), dtype=np.float64)
        counts = np.zeros(n_clusters, dtype=np.int64)
        np.add.at(sums, labels, e.astype(np.float64))
        np.add.at(counts, labels, 1)
        # empty clusters keep their previous centroid
        nz = counts > 0
        cent[nz] = normalize_rows((sums[nz] / counts[nz, None]).astype(np.float32))
    return cent.astype(np.float32), labels


def dist_to_centroid(
    e: npt.NDArray[np.float32],
    cent: npt.NDArray[np.float32],
    labels: npt.NDArray[np.int32],
) -> npt.NDArray[np.float32]:
    """cosine_dist_to_cent = 1 - cos(e_i, centroid[label_i]) (kmeans :260+)."""
    e = normalize_rows(e)
    cos = np.sum(e * cent[labels], axis=1)
    return (1.0 - cos).astype(np.float32)
