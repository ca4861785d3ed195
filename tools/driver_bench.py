"""End-to-end split-driver throughput (stage machinery + kernels).

bench.py measures the device hot path with inputs resident in HBM (the
BASELINE contract); this measures the WHOLE split() driver — synthetic
raw-NV12 corpus on disk -> download -> fixed-stride -> transcode ->
frame extraction -> embedding -> writer — single process, and prints the
per-stage StageTimer breakdown so orchestration cost is visible next to
the kernel cost.  Evidence for SURVEY §8b (the stage/runner machinery is
not the bottleneck story, quantified)."""
import argparse
import json
import pathlib
import shutil
import sys
import tempfile
import time

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))

from cosmos_curate_amd.core.interfaces import SequentialRunner  # noqa: E402
from cosmos_curate_amd.pipelines.video.splitting_pipeline import (  # noqa: E402
    _setup_parser,
    split,
)
from cosmos_curate_amd.pipelines.video.utils import raw_backend  # noqa: E402


def main(n_videos: int = 6, secs: int = 10) -> None:
    # raw NV12 is ~93 MB/s of video: keep the corpus small and in shm
    base = "/dev/shm" if pathlib.Path("/dev/shm").is_dir() else None
    tmp = pathlib.Path(tempfile.mkdtemp(prefix="drvbench_", dir=base))
    try:
        inp = tmp / "in"
        inp.mkdir()
        fps, h, w = 30, 1088, 1920
        for i in range(n_videos):
            (inp / f"v{i:03d}.nv12").write_bytes(
                raw_backend.make_synthetic_clip(fps * secs, h, w, fps, seed=i)
            )
        out = tmp / "out"
        p = argparse.ArgumentParser()
        _setup_parser(p)
        args = p.parse_args([
            "--input-video-path", str(inp), "--output-clip-path", str(out),
        ])
        t0 = time.perf_counter()
        summary = split(args, runner=SequentialRunner())
        dt = time.perf_counter() - t0
        n_clips = summary["num_clips"]
        print(json.dumps({
            "workload": f"split() end-to-end, {n_videos}x{secs}s 1080p30 raw-NV12, "
            "1 process (SequentialRunner)",
            "clips": n_clips,
            "embedded": summary["num_clips_with_embeddings"],
            "errors": summary["num_clips_with_errors"],
            "seconds": round(dt, 2),
            "clips_per_s": round(n_clips / dt, 2),
        }))
        for name, stats in sorted(summary.get("stage_perf", {}).items()):
            print(f"  {name}: {stats}")
    finally:
        shutil.rmtree(tmp, ignore_errors=True)


if __name__ == "__main__":
    main()
