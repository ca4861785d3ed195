"""Dedup HIP kernel parity + end-to-end semdedup on MI355X."""

import numpy as np
import pytest
import torch

from cosmos_curate_amd.pipelines.video.dedup import semdedup as sd
from oracle import semdedup as osd

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("m,d", [(1, 64), (63, 64), (64, 512), (129, 512), (1000, 512), (4097, 768)])
def test_pairwise_kernel_vs_oracle(m, d):
    rng = np.random.default_rng(m * 7 + d)
    e = rng.normal(size=(m, d)).astype(np.float32)
    maxv_o, argi_o = osd.pairwise_max_earlier(e)
    maxv, argi = sd.pairwise_max_earlier(torch.from_numpy(e).cuda())
    np.testing.assert_allclose(maxv.cpu().numpy(), maxv_o, atol=2e-5)
    # argmax may differ only where two scores are within fp tolerance
    am = argi.cpu().numpy()
    mismatch = am != argi_o
    if mismatch.any():
        en = osd.normalize_rows(e)
        for j in np.nonzero(mismatch)[0]:
            a, b = am[j], argi_o[j]
            va = float(en[a] @ en[j])
            vb = float(en[b] @ en[j])
            assert abs(va - vb) < 5e-5, (j, a, b, va, vb)


def test_pairwise_kernel_exact_duplicates():
    rng = np.random.default_rng(5)
    base = rng.normal(size=(100, 512)).astype(np.float32)
    e = np.concatenate([base, base[:40]])  # 40 exact duplicates at the end
    maxv, argi = sd.pairwise_max_earlier(torch.from_numpy(e).cuda())
    maxv = maxv.cpu().numpy()
    argi = argi.cpu().numpy()
    np.testing.assert_allclose(maxv[100:], 1.0, atol=1e-5)
    np.testing.assert_array_equal(argi[100:], np.arange(40))
    kept = maxv <= (1.0 - 0.01)
    assert kept[:100].all() or (~kept[:100]).sum() <= 2  # near-dup noise only
    assert not kept[100:].any()


def test_semdedup_end_to_end_gpu():
    rng = np.random.default_rng(1)
    base = osd.normalize_rows(rng.normal(size=(500, 512)).astype(np.float32))
    dup_idx = rng.choice(500, size=120, replace=False)
    e = np.concatenate([base, base[dup_idx]])
    perm = rng.permutation(len(e))
    e = e[perm]
    cfg = sd.SemDedupConfig(n_clusters=8, n_iters=10, eps=0.01)
    out = sd.semdedup(torch.from_numpy(e).cuda(), cfg)
    assert out["total"] == len(e)
    # every duplicate pair loses exactly one member
    assert out["kept"] == len(e) - 120


def test_pairwise_kernel_full_size_property():
    """BASELINE-scale property (config #5 cluster shard, ~100k rows):
    planted exact duplicates are found with score 1 pointing at the
    earliest copy, and all planted rows prune at eps=0.01."""
    rng = np.random.default_rng(99)
    m, d = 100_000, 512
    e = rng.normal(size=(m, d)).astype(np.float32)
    dup_src = rng.choice(m - 10_000, size=500, replace=False)
    dup_dst = m - 10_000 + np.arange(500) * 20  # spread through the tail
    e[dup_dst] = e[dup_src]
    maxv, argi = sd.pairwise_max_earlier(torch.from_numpy(e).cuda())
    maxv = maxv.cpu().numpy()
    argi = argi.cpu().numpy()
    np.testing.assert_allclose(maxv[dup_dst], 1.0, atol=2e-5)
    np.testing.assert_array_equal(argi[dup_dst], dup_src)
    # non-planted rows: random 512-d gaussians are near-orthogonal, so
    # their max cosine stays far below the prune threshold
    non_planted = m - 10_000 + np.arange(500) * 20 + 1
    assert (maxv[non_planted] < 0.9).all()
    kept = maxv <= 0.99
    assert not kept[dup_dst].any()
    assert kept[non_planted].all()


def test_dedup_pipeline_cli(tmp_path):
    """Standalone dedup pipeline (reference dedup_pipeline.py shape):
    embeddings parquet in -> results parquet + summary out, planted
    near-duplicates pruned."""
    import json

    import pyarrow as pa
    import pyarrow.parquet as pq
    import torch

    from cosmos_curate_amd.pipelines.video.dedup_pipeline import cli_run_dedup

    rng = np.random.default_rng(12)
    m, d = 600, 512
    emb = rng.normal(size=(m, d)).astype(np.float32)
    emb[:60] = emb[60:120] + 0.001 * rng.normal(size=(60, d)).astype(np.float32)
    ed = tmp_path / "clip_embd"
    ed.mkdir()
    pq.write_table(
        pa.table({"id": [f"c{i}" for i in range(m)],
                  "embedding": [e for e in emb]}),
        ed / "chunk_0.parquet",
    )
    out = tmp_path / "dedup"
    summary = cli_run_dedup([
        "--input-embeddings-path", str(tmp_path),
        "--output-dedup-path", str(out),
        "--n-clusters", "8", "--eps", "0.05",
    ])
    assert summary["num_embeddings"] == m
    assert summary["num_removed"] >= 55  # the planted near-dupes
    t = pq.read_table(out / "dedup_results.parquet")
    assert t.num_rows == m and set(t.column_names) == {"id", "cluster", "keep"}
    disk = json.loads((out / "summary.json").read_text())
    assert disk["num_kept"] == summary["num_kept"]
