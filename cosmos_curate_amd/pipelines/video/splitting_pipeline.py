"""Split pipeline driver — "Split-Transcode-Filter-Annotate".

Mirror of /root/reference/cosmos_curate/pipelines/video/
splitting_pipeline.py, hot-path subset: ``build_input_data`` (:187-260 —
one SplitPipeTask per input video, session_id = video path),
``_assemble_stages`` (:333-885 — ordered stage builders), ``split``/
``_split`` (:887-1020), and the flag subset covering the split/embed path
(the reference registers ~200 flags, :1025-2246; SURVEY.md §2 "Pipeline
driver": KEEP skeleton).

Stage order (reference _assemble_stages):
    VideoDownloader
    FixedStrideExtractorStage | VideoFrameExtraction+TransNetV2
    ClipTranscodingStage            (fans out via chunk_tasks)
    ClipFrameExtractionStage        (GPU decode->sample->resize)
    ClipFrameCreationStage
    ClipEmbeddingStage              (CLIP ViT-B/32, MFMA path)
    ClipWriterStage
"""

from __future__ import annotations

import argparse
import pathlib
import time

from cosmos_curate_amd.core.interfaces import (
    CuratorStage,
    CuratorStageSpec,
    RunnerInterface,
    run_pipeline,
)
from cosmos_curate_amd.pipelines.video.clipping.clip_extraction_stages import (
    ClipTranscodingStage,
    FixedStrideExtractorStage,
)
from cosmos_curate_amd.pipelines.video.clipping.clip_frame_extraction_stages import (
    ClipFrameExtractionStage,
)
from cosmos_curate_amd.pipelines.video.clipping.transnetv2_extraction_stages import (
    TransNetV2ClipExtractionStage,
    VideoFrameExtractionStage,
)
from cosmos_curate_amd.pipelines.video.embedding.clip_stages import (
    ClipEmbeddingStage,
    ClipFrameCreationStage,
)
from cosmos_curate_amd.pipelines.video.read_write.download_stages import VideoDownloader
from cosmos_curate_amd.pipelines.video.read_write.metadata_writer_stage import (
    ClipWriterStage,
    write_split_summary,
)
from cosmos_curate_amd.pipelines.video.utils.data_model import SplitPipeTask, Video
from cosmos_curate_amd.pipelines.video.utils.decoder_utils import FrameExtractionPolicy

VIDEO_SUFFIXES = {".mp4", ".mov", ".mkv", ".nv12", ".bin"}


def extract_multi_cam_split_tasks(
    sessions_prefix: pathlib.Path,
    primary_camera_keyword: str,
    limit: int = 0,
) -> list[SplitPipeTask]:
    """One task per UUID-named session dir, primary camera first
    (video_pipe_input.py:238-283 semantics, local filesystem)."""
    import uuid as uuid_mod

    def is_uuid(name: str) -> bool:
        try:
            uuid_mod.UUID(name)
        except ValueError:
            return False
        return True

    tasks: list[SplitPipeTask] = []
    for sess in sorted(p for p in sessions_prefix.iterdir()
                       if p.is_dir() and is_uuid(p.name)):
        vids = sorted(p for p in sess.rglob("*")
                      if p.is_file() and p.suffix.lower() in VIDEO_SUFFIXES)
        if not vids:
            continue
        # primary camera (path contains the keyword) goes to slot 0
        vids.sort(key=lambda p: (primary_camera_keyword not in str(p), str(p)))
        tasks.append(SplitPipeTask(
            session_id=sess.name,
            videos=[Video(input_video=v,
                          relative_path=str(v.relative_to(sessions_prefix)))
                    for v in vids],
        ))
        if limit > 0 and len(tasks) >= limit:
            break
    return tasks


def build_input_data(args: argparse.Namespace) -> list[SplitPipeTask]:
    """One task per input video or multicam session
    (splitting_pipeline.py:187-260 / :213-224)."""
    root = pathlib.Path(args.input_video_path)
    if getattr(args, "multi_cam", False):
        if args.splitting_algorithm != "fixed-stride":
            msg = "--multi-cam requires fixed-stride splitting (reference :213)"
            raise ValueError(msg)
        return extract_multi_cam_split_tasks(
            root, args.primary_camera_keyword, args.limit)
    if getattr(args, "input_video_list_json_path", None):
        import json as json_mod

        listed = json_mod.loads(
            pathlib.Path(args.input_video_list_json_path).read_text())
        files = [pathlib.Path(p) for p in listed]
    else:
        files = sorted(
            p for p in root.rglob("*")
            if p.is_file() and p.suffix.lower() in VIDEO_SUFFIXES
        )
    if args.limit > 0:
        files = files[: args.limit]
    return [
        SplitPipeTask(
            session_id=str(p),
            videos=[Video(
                input_video=p,
                relative_path=str(p.relative_to(root)) if root in p.parents
                else p.name,
            )],
        )
        for p in files
    ]


def _assemble_stages(args: argparse.Namespace) -> list[CuratorStage | CuratorStageSpec]:
    """Ordered stage list (splitting_pipeline.py:333-885)."""
    stages: list[CuratorStage | CuratorStageSpec] = [
        VideoDownloader(args.input_video_path, verbose=args.verbose, log_stats=True),
    ]
    if args.splitting_algorithm == "fixed-stride":
        stages.append(
            FixedStrideExtractorStage(
                clip_len_s=args.fixed_stride_split_duration,
                clip_stride_s=args.fixed_stride_split_duration,
                min_clip_length_s=args.fixed_stride_min_clip_length,
                limit_clips=args.limit_clips,
                log_stats=True,
            )
        )
    elif args.splitting_algorithm == "transnetv2":
        stages.append(VideoFrameExtractionStage(log_stats=True))
        stages.append(
            TransNetV2ClipExtractionStage(
                threshold=args.transnetv2_threshold,
                limit_clips=args.limit_clips,
                log_stats=True,
            )
        )
    else:
        msg = f"Unknown splitting algorithm {args.splitting_algorithm}"
        raise ValueError(msg)
    stages.append(
        ClipTranscodingStage(num_clips_per_chunk=args.num_clips_per_chunk, log_stats=True)
    )
    if args.generate_embeddings or args.aesthetic_threshold is not None:
        fps_targets: list[float | int] = [args.target_clip_fps]
        if args.aesthetic_threshold is not None and 1.0 not in fps_targets:
            fps_targets.append(1.0)  # aesthetics samples at 1 fps (builder default)
        res = args.clip_extraction_target_res
        stages.append(
            ClipFrameExtractionStage(
                extraction_policies=(FrameExtractionPolicy.sequence,),
                target_fps=sorted(fps_targets),
                target_res=(res, res) if res > 0 else (-1, -1),
                log_stats=True,
            )
        )
    if args.aesthetic_threshold is not None:
        from cosmos_curate_amd.pipelines.video.filtering.aesthetics.aesthetic_filter_stages import (
            AestheticFilterStage,
        )

        stages.append(
            AestheticFilterStage(score_threshold=args.aesthetic_threshold, log_stats=True)
        )
    if args.motion_filter != "disable":
        # MV extraction itself sits behind the rocDecode seam (DESIGN.md
        # §4): clips without decoded_motion_data record a loud per-clip
        # error; the stage order and filter semantics mirror the
        # reference (:1340-1376)
        from cosmos_curate_amd.pipelines.video.filtering.motion.motion_filter_stages import (
            MotionFilterStage,
        )

        stages.append(MotionFilterStage(
            global_mean_threshold=args.motion_global_mean_threshold,
            per_patch_min_256_threshold=args.motion_per_patch_min_256_threshold,
            score_only=args.motion_filter == "score-only",
            log_stats=True,
        ))
    if args.generate_embeddings:
        res = args.clip_extraction_target_res
        stages.append(ClipFrameCreationStage(
            target_fps=args.target_clip_fps,
            target_res=(res, res) if res > 0 else (-1, -1),
            log_stats=True,
        ))
        stages.append(ClipEmbeddingStage(log_stats=True))
    stages.append(ClipWriterStage(args.output_clip_path, log_stats=True))
    return stages


def _setup_parser(parser: argparse.ArgumentParser) -> None:
    """Flag subset of splitting_pipeline.py:1025-2246."""
    parser.add_argument("--input-video-path", required=True)
    parser.add_argument("--output-clip-path", required=True)
    parser.add_argument(
        "--splitting-algorithm", default="fixed-stride",
        choices=["fixed-stride", "transnetv2"],
    )
    parser.add_argument("--fixed-stride-split-duration", type=float, default=10.0)
    parser.add_argument("--fixed-stride-min-clip-length",
                        "--fixed-stride-min-clip-length-s",  # reference spelling
                        dest="fixed_stride_min_clip_length",
                        type=float, default=10.0)
    parser.add_argument("--transnetv2-threshold", type=float, default=0.4)
    parser.add_argument("--limit", type=int, default=0)
    parser.add_argument("--limit-clips", type=int, default=0)
    parser.add_argument("--num-clips-per-chunk", type=int, default=32)
    parser.add_argument("--target-clip-fps", type=float, default=2.0)
    parser.add_argument("--generate-embeddings", action="store_true", default=True)
    parser.add_argument("--no-embeddings", "--no-generate-embeddings",
                        dest="generate_embeddings", action="store_false")
    parser.add_argument("--embedding-algorithm", default="clip")
    parser.add_argument("--aesthetic-threshold", type=float, default=None)
    parser.add_argument("--clip-extraction-target-res", type=int, default=224,
                        help="square target resolution for extracted frames; "
                        "-1 disables resize (reference flag; the embedder "
                        "resizes on device if needed)")
    parser.add_argument("--motion-filter", choices=["disable", "enable", "score-only"],
                        default="disable")
    parser.add_argument("--motion-global-mean-threshold", type=float, default=0.00098)
    parser.add_argument("--motion-per-patch-min-256-threshold", type=float,
                        default=0.000001)
    parser.add_argument("--multi-cam", action="store_true",
                        help="input path holds UUID-named session dirs, one "
                        "task per session (fixed-stride only, reference :213)")
    parser.add_argument("--primary-camera-keyword", default="front")
    parser.add_argument("--input-video-list-json-path", default=None,
                        help="JSON list of video paths instead of directory scan")
    parser.add_argument("--dry-run", action="store_true",
                        help="print the assembled stage list and exit")
    parser.add_argument("--stage-save", default=None,
                        help="save per-stage input/output task pickles here "
                        "(the --stage-save/replay/compare harness)")
    parser.add_argument("--verbose", action="store_true")
    parser.add_argument("--perf-profile", action="store_true")


def split(args: argparse.Namespace, runner: RunnerInterface | None = None) -> dict:
    """Run the split pipeline end-to-end (splitting_pipeline.py:887-1020)."""
    t0 = time.perf_counter()
    input_tasks = build_input_data(args)
    stages = _assemble_stages(args)
    if getattr(args, "dry_run", False):
        for st in stages:
            inner = st.stage if hasattr(st, "stage") else st
            print(type(inner).__name__)
        return {"dry_run": True, "num_input_videos": len(input_tasks),
                "num_stages": len(stages)}
    save_cfg = None
    if getattr(args, "stage_save", None):
        from cosmos_curate_amd.core.utils.stage_replay import StageSaveConfig

        save_cfg = StageSaveConfig(output_path=args.stage_save)
    output_tasks = run_pipeline(
        input_tasks, stages, runner=runner, stage_save_config=save_cfg
    )
    elapsed = time.perf_counter() - t0
    return write_split_summary(
        args.output_clip_path,
        output_tasks,
        embedding_algorithm=args.embedding_algorithm,
        pipeline_run_time_s=elapsed,
    )


def cli_run_split(argv: list[str] | None = None) -> dict:
    parser = argparse.ArgumentParser("split")
    _setup_parser(parser)
    args = parser.parse_args(argv)
    summary = split(args)
    print(
        f"split: {summary['num_input_videos']} videos -> {summary['num_clips']} clips, "
        f"{summary['num_clips_with_embeddings']} embedded, "
        f"{summary['num_clips_with_errors']} errors"
    )
    return summary


if __name__ == "__main__":
    cli_run_split()
