"""Semantic-dedup pipeline driver (reference dedup_pipeline.py:175-300).

Standalone pipeline, like upstream: reads the embeddings parquet the
split pipeline wrote (`<split-out>/<alg>_embd/chunk_*.parquet`, id +
embedding columns), runs RCCL-k-means + per-cluster max-cosine pruning
(dedup/semdedup.py — the cuML KMeansMG / RAFT-NCCL replacement,
SURVEY.md §8f row 3), and writes

    <out>/dedup_results.parquet   id, cluster, keep
    <out>/summary.json            counts + parameters

Multi-GPU: launched one process per GPU (torchrun / WORLD_SIZE env like
bench.py); ranks share the k-means all-reduce over RCCL and split
clusters round-robin, then keep-masks are all-gathered to rank 0.

No CPU fallback: the pairwise kernel requires a GPU (hotpath gate); a
GPU-less invocation fails loudly.
"""

from __future__ import annotations

import argparse
import json
import os
import pathlib
import time

import numpy as np
import torch

from cosmos_curate_amd.pipelines.video.dedup.semdedup import (
    SemDedupConfig,
    combine_keep_masks,
    semdedup,
)


def load_embeddings(path: pathlib.Path) -> tuple[list[str], np.ndarray]:
    """ids + (m, d) f32 from the split writer's parquet chunks."""
    import pyarrow.parquet as pq

    files = sorted(path.glob("*_embd/*.parquet")) or sorted(path.glob("*.parquet"))
    if not files:
        msg = f"no embeddings parquet under {path}"
        raise FileNotFoundError(msg)
    ids: list[str] = []
    rows: list[np.ndarray] = []
    for f in files:
        t = pq.read_table(f)
        ids.extend(t.column("id").to_pylist())
        rows.extend(np.asarray(v, dtype=np.float32) for v in t.column("embedding").to_pylist())
    return ids, np.stack(rows)


def _setup_parser(p: argparse.ArgumentParser) -> None:
    p.add_argument("--input-embeddings-path", required=True)
    p.add_argument("--output-dedup-path", required=True)
    p.add_argument("--n-clusters", type=int, default=100)
    p.add_argument("--n-iters", type=int, default=20)
    p.add_argument("--eps", type=float, default=0.01,
                   help="prune a row when max-cosine to an earlier row > 1-eps")


def dedup(args: argparse.Namespace) -> dict:
    world = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    group = None
    if world > 1:
        torch.distributed.init_process_group("nccl")
        group = torch.distributed.group.WORLD
    torch.cuda.set_device(local_rank)

    ids, emb = load_embeddings(pathlib.Path(args.input_embeddings_path))
    e = torch.from_numpy(emb).cuda()
    cfg = SemDedupConfig(n_clusters=args.n_clusters, n_iters=args.n_iters,
                         eps=args.eps)
    t0 = time.perf_counter()
    out = semdedup(e, cfg, process_group=group)
    keep = combine_keep_masks(out["keep_mask"], group)
    dt = time.perf_counter() - t0

    summary = {
        "num_embeddings": len(ids),
        "num_kept": int(keep.sum().item()),
        "num_removed": int((~keep).sum().item()),
        "n_clusters": cfg.n_clusters,
        "eps": cfg.eps,
        "seconds": round(dt, 3),
        "world_size": world,
    }
    if rank == 0:
        outdir = pathlib.Path(args.output_dedup_path)
        outdir.mkdir(parents=True, exist_ok=True)
        import pyarrow as pa
        import pyarrow.parquet as pq

        pq.write_table(
            pa.table({
                "id": ids,
                "cluster": out["labels"].cpu().numpy().astype(np.int32),
                "keep": keep.cpu().numpy(),
            }),
            outdir / "dedup_results.parquet",
        )
        (outdir / "summary.json").write_text(json.dumps(summary, indent=1))
    if group is not None:
        torch.distributed.destroy_process_group()
    return summary


def cli_run_dedup(argv: list[str] | None = None) -> dict:
    p = argparse.ArgumentParser("dedup")
    _setup_parser(p)
    summary = dedup(p.parse_args(argv))
    print(f"dedup: {summary['num_embeddings']} embeddings -> "
          f"{summary['num_kept']} kept, {summary['num_removed']} removed")
    return summary


if __name__ == "__main__":
    cli_run_dedup()
