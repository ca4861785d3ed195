"""Semantic dedup: oracle properties + product kmeans parity (CPU)."""

import numpy as np
import pytest
import torch

from cosmos_curate_amd.pipelines.video.dedup import semdedup as sd
from oracle import semdedup as osd


def test_oracle_pairwise_semantics():
    # three exact duplicates of row 0 at positions 0, 2, 4
    e = np.array(
        [[1, 0, 0, 0], [0, 1, 0, 0], [1, 0, 0, 0], [0, 0, 1, 0], [1, 0, 0, 0]],
        dtype=np.float32,
    )
    maxv, argi = osd.pairwise_max_earlier(e)
    assert maxv[0] == 0.0 and argi[0] == 0
    np.testing.assert_allclose(maxv[2], 1.0, atol=1e-6)
    assert argi[2] == 0  # first occurrence
    np.testing.assert_allclose(maxv[4], 1.0, atol=1e-6)
    assert argi[4] == 0  # tie (rows 0 and 2) resolves to the earliest
    kept = osd.kept_mask(maxv, eps=0.01)
    np.testing.assert_array_equal(kept, [True, True, False, True, False])


def test_oracle_pairwise_random_bruteforce():
    rng = np.random.default_rng(3)
    e = rng.normal(size=(67, 16)).astype(np.float32)
    maxv, argi = osd.pairwise_max_earlier(e)
    en = osd.normalize_rows(e)
    sims = np.clip(en @ en.T, -1, 1)
    for j in range(1, len(e)):
        col = sims[:j, j]
        assert maxv[j] == pytest.approx(col.max(), abs=1e-6)
        assert argi[j] == int(np.argmax(col))


def test_product_kmeans_matches_oracle_single_rank():
    rng = np.random.default_rng(11)
    e = rng.normal(size=(300, 32)).astype(np.float32)
    cent_o, labels_o = osd.kmeans(e, n_clusters=8, n_iters=10, seed=42)
    cent_p, labels_p = sd.kmeans_fit(torch.from_numpy(e), 8, 10, seed=42)
    np.testing.assert_array_equal(labels_p.numpy(), labels_o)
    np.testing.assert_allclose(cent_p.numpy(), cent_o, atol=1e-5)


def test_kmeans_clusters_separable_data():
    rng = np.random.default_rng(0)
    centers = osd.normalize_rows(rng.normal(size=(4, 64)).astype(np.float32))
    e = np.concatenate(
        [c + 0.05 * rng.normal(size=(50, 64)).astype(np.float32) for c in centers]
    )
    cent, labels = sd.kmeans_fit(torch.from_numpy(e), 4, 15, seed=1)
    groups = labels.numpy().reshape(4, 50)
    for g in groups:  # each true cluster lands in one k-means cluster
        assert len(np.unique(g)) == 1
    assert len(np.unique(groups[:, 0])) == 4
