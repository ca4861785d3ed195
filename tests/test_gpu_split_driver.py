"""Full split pipeline with embeddings on MI355X (the flagship path)."""

import argparse
import json

import numpy as np
import pytest

from cosmos_curate_amd.core.interfaces import SequentialRunner
from cosmos_curate_amd.pipelines.video.splitting_pipeline import _setup_parser, split
from cosmos_curate_amd.pipelines.video.utils import raw_backend

pytestmark = pytest.mark.gpu


def test_split_end_to_end_with_embeddings(tmp_path):
    inp = tmp_path / "in"
    inp.mkdir()
    for i in range(2):
        raw = raw_backend.make_synthetic_clip(600, 128, 192, 30, seed=10 + i)
        (inp / f"v{i}.nv12").write_bytes(raw)
    out = tmp_path / "out"
    p = argparse.ArgumentParser()
    _setup_parser(p)
    args = p.parse_args(
        ["--input-video-path", str(inp), "--output-clip-path", str(out)]
    )
    summary = split(args, runner=SequentialRunner())
    # 20 s videos, 10 s stride -> 2 clips each
    assert summary["num_input_videos"] == 2
    assert summary["num_clips"] == 4
    assert summary["num_clips_with_embeddings"] == 4
    assert summary["num_clips_with_errors"] == 0
    assert summary["clips_per_second"] > 0
    # embeddings parquet readable and unit-norm
    import pyarrow.parquet as pq

    chunks = list((out / "clip_embd").glob("*.parquet"))
    assert chunks
    rows = 0
    for c in chunks:
        t = pq.read_table(c)
        rows += t.num_rows
        for emb in t.column("embedding").to_pylist():
            v = np.array(emb, dtype=np.float32)
            assert v.shape == (512,)
            np.testing.assert_allclose(np.linalg.norm(v), 1.0, atol=1e-3)
    assert rows == 4
    disk = json.loads((out / "summary.json").read_text())
    assert "ClipEmbeddingStage" in disk["stage_perf"]
