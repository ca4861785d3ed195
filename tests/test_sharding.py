"""Sharding pipeline: bin spec, greedy packer, tar round-trip (CPU)."""

import json
import tarfile

import numpy as np

from cosmos_curate_amd.pipelines.video import sharding_pipeline as shp


def sample(uuid, nbytes=1000, w=1920, h=1080, fps=30.0, frames=300):
    return shp.ClipSample(uuid=uuid, payload=bytes(nbytes), width=w, height=h,
                          framerate=fps, num_frames=frames,
                          metadata={"uuid": uuid, "n": nbytes})


def test_bin_key_classes():
    assert shp.bin_key(sample("a")) == "res1080_ar16-9_5-10s"
    assert shp.bin_key(sample("b", w=640, h=480, frames=60)) == "res480_ar4-3_0-2s"
    assert shp.bin_key(sample("c", w=1080, h=1920)) == "res1080_ar9-16_5-10s"
    assert shp.bin_key(sample("d", fps=0.0)) is None


def test_greedy_packer_semantics():
    ss = [sample(f"s{i}", nbytes=400) for i in range(7)]
    packs = list(shp.group_samples_by_size(ss, 1000))
    assert [len(p) for p in packs] == [2, 2, 2, 1]
    # small tail dropped when requested (min_clips_per_tar=2)
    packs = list(shp.group_samples_by_size(ss, 1000, drop_small_shards=True))
    assert [len(p) for p in packs] == [2, 2, 2]
    # oversize sample still gets its own tar
    packs = list(shp.group_samples_by_size([sample("big", nbytes=5000)], 1000))
    assert [len(p) for p in packs] == [1]


def test_write_and_roundtrip(tmp_path):
    ss = [sample(f"u{i}", nbytes=500 + i) for i in range(5)]
    ss.append(sample("v0", w=640, h=480))
    index = shp.write_webdataset_shards(ss, str(tmp_path), target_size_bytes=1200)
    disk = json.loads((tmp_path / "shard_index.json").read_text())
    assert disk == index
    assert "res1080_ar16-9_5-10s" in index and "res480_ar4-3_5-10s" in index
    total = 0
    for label, shards in index.items():
        for s in shards:
            with tarfile.open(tmp_path / s["tar"]) as tar:
                names = tar.getnames()
                bins = [n for n in names if n.endswith(".bin")]
                metas = [n for n in names if n.endswith(".json")]
                assert len(bins) == len(metas) == s["clips"]
                for m in metas:
                    meta = json.loads(tar.extractfile(m).read())
                    payload = tar.extractfile(meta["uuid"] + ".bin").read()
                    assert len(payload) == meta["n"]
            total += s["clips"]
    assert total == 6


def test_shard_from_split_output(tmp_path):
    """split (raw corpus, no embeddings) -> shard end-to-end on CPU."""
    import argparse

    from cosmos_curate_amd.core.interfaces import SequentialRunner
    from cosmos_curate_amd.pipelines.video.splitting_pipeline import _setup_parser, split
    from cosmos_curate_amd.pipelines.video.utils import raw_backend

    inp = tmp_path / "in"
    inp.mkdir()
    (inp / "v.nv12").write_bytes(raw_backend.make_synthetic_clip(600, 64, 96, 30, seed=1))
    out = tmp_path / "split"
    p = argparse.ArgumentParser()
    _setup_parser(p)
    args = p.parse_args(["--input-video-path", str(inp), "--output-clip-path",
                         str(out), "--no-embeddings"])
    split(args, runner=SequentialRunner())

    index = shp.cli_run_shard([
        "--input-clip-path", str(out), "--output-shard-path", str(tmp_path / "shards"),
        "--target-shard-size-mb", "1",
    ])
    total = sum(s["clips"] for shards in index.values() for s in shards)
    assert total == 2  # 20 s video -> two 10 s clips


def test_shard_consumes_dedup_results(tmp_path):
    """split -> dedup results -> shard: pruned clips excluded
    (video_pipe_input.py:514-560 semantics)."""
    import argparse

    import pyarrow as pa
    import pyarrow.parquet as pq

    from cosmos_curate_amd.core.interfaces import SequentialRunner
    from cosmos_curate_amd.pipelines.video.splitting_pipeline import _setup_parser, split
    from cosmos_curate_amd.pipelines.video.utils import raw_backend

    inp = tmp_path / "in"
    inp.mkdir()
    (inp / "v.nv12").write_bytes(raw_backend.make_synthetic_clip(600, 64, 96, 30, seed=1))
    out = tmp_path / "split"
    p = argparse.ArgumentParser()
    _setup_parser(p)
    split(p.parse_args(["--input-video-path", str(inp), "--output-clip-path",
                        str(out), "--no-embeddings"]), runner=SequentialRunner())
    metas = sorted((out / "metas" / "v0").glob("*.json"))
    ids = [json.loads(m.read_text())["uuid"] for m in metas]
    assert len(ids) == 2
    dd = tmp_path / "dedup"
    dd.mkdir()
    pq.write_table(pa.table({"id": ids, "cluster": [0, 0],
                             "keep": [True, False]}),
                   dd / "dedup_results.parquet")
    index = shp.cli_run_shard([
        "--input-clip-path", str(out), "--output-shard-path", str(tmp_path / "sh"),
        "--target-shard-size-mb", "1",
        "--input-semantic-dedup-path", str(dd),
    ])
    total = sum(s["clips"] for shards in index.values() for s in shards)
    assert total == 1  # the pruned clip is excluded


def test_multicam_split_to_shard(tmp_path):
    """Multicam split -> shard: cam-prefixed sample ids resolve payloads
    and match dedup ids (previously every multicam clip was
    silently skipped because the loader looked up clips/<bare-uuid>.bin)."""
    import argparse
    import uuid as uuid_mod

    import pyarrow as pa
    import pyarrow.parquet as pq

    from cosmos_curate_amd.core.interfaces import SequentialRunner
    from cosmos_curate_amd.pipelines.video.splitting_pipeline import _setup_parser, split
    from cosmos_curate_amd.pipelines.video.utils import raw_backend

    root = tmp_path / "sessions"
    sid = str(uuid_mod.uuid4())
    for cam in ["front", "rear"]:
        d = root / sid / cam
        d.mkdir(parents=True)
        (d / "v.nv12").write_bytes(
            raw_backend.make_synthetic_clip(600, 32, 48, 30, seed=3))
    out = tmp_path / "split"
    p = argparse.ArgumentParser()
    _setup_parser(p)
    split(p.parse_args([
        "--input-video-path", str(root), "--output-clip-path", str(out),
        "--multi-cam", "--no-embeddings",
    ]), runner=SequentialRunner())

    samples = shp.load_samples_from_split_output(str(out))
    # 2 cams x 2 clips, all loaded (cam-prefixed payload names resolve)
    assert len(samples) == 4
    assert all(s.uuid.startswith("cam") for s in samples)
    assert {s.metadata["camera_index"] for s in samples} == {0, 1}

    # dedup results keyed by the same cam-prefixed ids: prune one camera's
    # clip and verify exactly that sample drops out
    ids = sorted(s.uuid for s in samples)
    dd = tmp_path / "dedup"
    dd.mkdir()
    pq.write_table(pa.table({"id": ids, "cluster": [0] * 4,
                             "keep": [False, True, True, True]}),
                   dd / "dedup_results.parquet")
    kept = shp.filter_samples_by_semantic_dedup(samples, str(dd))
    assert sorted(s.uuid for s in kept) == ids[1:]

    index = shp.write_webdataset_shards(kept, str(tmp_path / "shards"),
                                        target_size_bytes=1 << 20)
    total = sum(s["clips"] for shards in index.values() for s in shards)
    assert total == 3
