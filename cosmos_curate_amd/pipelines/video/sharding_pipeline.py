"""Sharding pipeline — pack curated clips into webdataset tars.

Mirror of /root/reference/cosmos_curate/pipelines/video/
sharding_pipeline.py: samples bin by
resolution/aspect/duration (``_group_samples_by_bin`` :69-88, the
``for_standard_video_datasets`` bin spec), greedy size-packing into tars
with a target byte size and a small-tail drop rule
(``_group_samples_by_size`` :90-114), tar members = clip payload +
metadata json keyed by clip uuid (webdataset convention).

Input is a split-pipeline output tree (clips/ + metas/v0/, the
ClipWriterStage layout); output is

    <out>/<bin>/shard_00000.tar ... + shard_index.json
"""

from __future__ import annotations

import argparse
import dataclasses
import io
import json
import pathlib
import tarfile
from collections import defaultdict
from collections.abc import Generator, Iterable

MIN_CLIPS_PER_TAR_DEFAULT = 2

# our bin spec analog of dimensions.ResolutionAspectRatioFramesBinsSpec
# .for_standard_video_datasets(): min-side resolution classes, nearest
# canonical aspect, duration buckets in seconds.
RESOLUTION_BINS = (240, 360, 480, 720, 1080, 2160)
ASPECT_BINS = (("16-9", 16 / 9), ("4-3", 4 / 3), ("1-1", 1.0), ("9-16", 9 / 16))
DURATION_BINS = ((2, "0-2s"), (5, "2-5s"), (10, "5-10s"), (30, "10-30s"),
                 (10 ** 9, "30s+"))


@dataclasses.dataclass
class ClipSample:
    """data_model.py:804-833 subset: one packable clip."""

    uuid: str
    payload: bytes
    width: int
    height: int
    framerate: float
    num_frames: int
    metadata: dict

    @property
    def num_bytes(self) -> int:
        return len(self.payload)

    @property
    def duration(self) -> float:
        return self.num_frames / self.framerate if self.framerate > 0 else 0.0


def bin_key(sample: ClipSample) -> str | None:
    """Resolution/aspect/duration bin label (sharding_pipeline.py:69-88)."""
    if sample.framerate <= 0 or sample.width <= 0 or sample.height <= 0:
        return None
    min_side = min(sample.width, sample.height)
    res = next((r for r in RESOLUTION_BINS if min_side <= r), RESOLUTION_BINS[-1])
    ratio = sample.width / sample.height
    aspect = min(ASPECT_BINS, key=lambda ab: abs(ab[1] - ratio))[0]
    dur = next(label for limit, label in DURATION_BINS if sample.duration <= limit)
    return f"res{res}_ar{aspect}_{dur}"


def group_samples_by_bin(samples: Iterable[ClipSample]) -> dict[str | None, list[ClipSample]]:
    out: dict[str | None, list[ClipSample]] = defaultdict(list)
    for s in samples:
        out[bin_key(s)].append(s)
    return out


def group_samples_by_size(
    samples: list[ClipSample],
    target_size_bytes: int,
    *,
    drop_small_shards: bool = False,
    min_clips_per_tar: int = MIN_CLIPS_PER_TAR_DEFAULT,
) -> Generator[list[ClipSample], None, None]:
    """Greedy packer (sharding_pipeline.py:90-114 semantics, incl. the
    drop-small-tail rule)."""
    current: list[ClipSample] = []
    size = 0
    for s in samples:
        if not current:
            current = [s]
            size = s.num_bytes
        elif size + s.num_bytes > target_size_bytes:
            yield current
            current = [s]
            size = s.num_bytes
        else:
            current.append(s)
            size += s.num_bytes
    if current and not (drop_small_shards and len(current) < min_clips_per_tar):
        yield current


def write_webdataset_shards(
    samples: Iterable[ClipSample],
    output_path: str,
    target_size_bytes: int = 256 * 1024 * 1024,
    *,
    drop_small_shards: bool = False,
    min_clips_per_tar: int = MIN_CLIPS_PER_TAR_DEFAULT,
) -> dict:
    """Bin + pack + write tars; returns the shard index (also on disk)."""
    out_root = pathlib.Path(output_path)
    index: dict[str, list[dict]] = {}
    for bkey, group in sorted(group_samples_by_bin(samples).items(), key=lambda kv: str(kv[0])):
        label = bkey if bkey is not None else "unbinned"
        bin_dir = out_root / label
        bin_dir.mkdir(parents=True, exist_ok=True)
        shards = []
        for i, pack in enumerate(
            group_samples_by_size(group, target_size_bytes, drop_small_shards=drop_small_shards, min_clips_per_tar=min_clips_per_tar)
        ):
            tar_path = bin_dir / f"shard_{i:05d}.tar"
            with tarfile.open(tar_path, "w") as tar:
                for s in pack:
                    payload_info = tarfile.TarInfo(f"{s.uuid}.bin")
                    payload_info.size = len(s.payload)
                    tar.addfile(payload_info, io.BytesIO(s.payload))
                    meta = json.dumps(s.metadata).encode()
                    meta_info = tarfile.TarInfo(f"{s.uuid}.json")
                    meta_info.size = len(meta)
                    tar.addfile(meta_info, io.BytesIO(meta))
            shards.append(
                {"tar": str(tar_path.relative_to(out_root)), "clips": len(pack),
                 "bytes": sum(s.num_bytes for s in pack)}
            )
        index[label] = shards
    out_root.mkdir(parents=True, exist_ok=True)
    (out_root / "shard_index.json").write_text(json.dumps(index, indent=1))
    return index


def load_samples_from_split_output(split_output: str) -> list[ClipSample]:
    """Read a ClipWriterStage tree (clips/ + metas/v0/) into samples."""
    root = pathlib.Path(split_output)
    samples = []
    for meta_path in sorted((root / "metas" / "v0").glob("*.json")):
        meta = json.loads(meta_path.read_text())
        # sample id = meta filename stem = ClipWriterStage's file id
        # (cam-prefixed for multicam, bare uuid otherwise); matches the
        # clips/<id>.bin payload name and the embedding-parquet "id"
        # column that dedup_results.parquet keys on
        sample_id = meta.get("id", meta_path.stem)
        payload_path = root / "clips" / f"{sample_id}.bin"
        if not payload_path.exists():
            continue
        payload = payload_path.read_bytes()
        # probe dimensions from the payload (raw header or mp4)
        from cosmos_curate_amd.pipelines.video.utils import raw_backend

        if raw_backend.is_raw_nv12(payload):
            n, h, w, fps = raw_backend.parse_header(payload)
        else:
            from cosmos_curate_amd.pipelines.video.utils.decoder_utils import (
                extract_video_metadata,
            )

            md = extract_video_metadata(payload)
            n, h, w, fps = md.num_frames, md.height, md.width, md.fps
        samples.append(
            ClipSample(uuid=sample_id, payload=payload, width=w, height=h,
                       framerate=fps, num_frames=n, metadata=meta)
        )
    return samples


def filter_samples_by_semantic_dedup(
    samples: list[ClipSample], dedup_path: str,
) -> list[ClipSample]:
    """Drop samples the dedup pipeline pruned
    (video_pipe_input.py:514-560 semantics over dedup_pipeline.py's
    dedup_results.parquet: keep == False rows are removed)."""
    import pathlib

    import pyarrow.parquet as pq

    files = sorted(pathlib.Path(dedup_path).rglob("dedup_results.parquet"))
    if not files:
        msg = f"no dedup_results.parquet under {dedup_path}"
        raise FileNotFoundError(msg)
    remove: set[str] = set()
    for f in files:
        t = pq.read_table(f, columns=["id", "keep"])
        for cid, keep in zip(t.column("id").to_pylist(),
                             t.column("keep").to_pylist()):
            if not keep:
                remove.add(str(cid))
    return [s for s in samples if str(s.uuid) not in remove]


def shard(args: argparse.Namespace) -> dict:
    """sharding_pipeline.py:194 driver shape."""
    samples = load_samples_from_split_output(args.input_clip_path)
    if getattr(args, "input_semantic_dedup_path", None):
        samples = filter_samples_by_semantic_dedup(
            samples, args.input_semantic_dedup_path)
    return write_webdataset_shards(
        samples, args.output_shard_path,
        target_size_bytes=args.target_shard_size_mb * 1024 * 1024,
        drop_small_shards=args.drop_small_shards,
        min_clips_per_tar=getattr(args, "min_clips_per_tar",
                                  MIN_CLIPS_PER_TAR_DEFAULT),
    )


def cli_run_shard(argv: list[str] | None = None) -> dict:
    p = argparse.ArgumentParser("shard")
    p.add_argument("--input-clip-path", required=True)
    p.add_argument("--output-shard-path", required=True)
    p.add_argument("--target-shard-size-mb", type=int, default=256)
    p.add_argument("--drop-small-shards", action="store_true")
    p.add_argument("--min-clips-per-tar", type=int, default=MIN_CLIPS_PER_TAR_DEFAULT)
    p.add_argument("--input-semantic-dedup-path", default=None,
                   help="dedup pipeline output dir; pruned clips are "
                   "excluded from the shards (reference "
                   "--input-semantic-dedup-path)")
    args = p.parse_args(argv)
    index = shard(args)
    total = sum(s["clips"] for shards in index.values() for s in shards)
    print(f"shard: {total} clips into {sum(len(v) for v in index.values())} tars, "
          f"{len(index)} bins")
    return index


if __name__ == "__main__":
    cli_run_shard()
