"""Semantic dedup — multi-GPU k-means + pairwise max-cosine pruning.

Mirror of /root/reference/cosmos_curate/pipelines/video/dedup/
(dedup_pipeline.py:237 ``dedup``, dedup_actor.py ``SemDedupActor``
kmeans :182-313 / dedup :315-460, raft_actor.py NCCL wiring :84-132),
rebuilt MI355X-native (BASELINE config #5):

- the cuML ``KMeansMG`` fit over RAFT/NCCL becomes torch spherical
  k-means with ``torch.distributed`` all-reduce of centroid sums+counts
  over RCCL/xGMI (the same collective cuML issues per iteration); the
  unique-id broadcast dance of raft_actor.py:84-119 collapses into the
  process group init;
- the cuPy tiled strict-upper-triangular scan (dedup :393-460) becomes
  one HIP kernel on exact-f32 MFMA (csrc/cc_dedup.hip,
  ``cc_pairwise_max_earlier``);
- "kept" semantics identical: row pruned iff max-cosine to an earlier
  (farther-from-centroid) row exceeds 1 - eps.

No CPU fallback: ``pairwise_max_earlier`` requires the HIP extension +
GPU.  ``kmeans`` is plain torch and runs wherever its tensors live —
the world_size-2 gloo test drives the collective path on CPU.
"""

from __future__ import annotations

import dataclasses

import numpy as np
import torch

from cosmos_curate_amd import hotpath


@dataclasses.dataclass
class SemDedupConfig:
    """dedup_actor.py:47-74 subset."""

    n_clusters: int = 100
    n_iters: int = 20
    eps: float = 0.01
    random_seed: int = 0x5EED


def kmeans_fit(
    emb: torch.Tensor,
    n_clusters: int,
    n_iters: int = 20,
    seed: int = 0x5EED,
    process_group=None,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Spherical k-means over (possibly rank-sharded) embeddings.

    Each rank holds a shard; per iteration, assignment is local and the
    centroid sums/counts are all-reduced (the KMeansMG collective,
    dedup_actor.py:197-232).  Deterministic: init centroids are sampled
    by the shared seed from rank 0's shard size convention — all ranks
    compute the same init when fed identical full data (single rank) or
    receive rank-0's init via broadcast (multi rank).

    Returns (centroids [k,d] unit rows, local labels [m]).
    """
    e = torch.nn.functional.normalize(emb.float(), dim=1, eps=1e-12)
    m, d = e.shape
    dist = torch.distributed if process_group is not None else None

    rng = np.random.default_rng(seed)
    if dist is None or dist.get_rank(process_group) == 0:
        init_idx = np.sort(rng.choice(m, size=n_clusters, replace=False))
        cent = e[torch.from_numpy(init_idx.astype(np.int64))].clone()
    else:
        cent = torch.empty((n_clusters, d), dtype=e.dtype, device=e.device)
    if dist is not None:
        dist.broadcast(cent, src=0, group=process_group)

    labels = torch.zeros(m, dtype=torch.int64, device=e.device)
    for _ in range(n_iters):
        sims = e @ cent.T
        labels = sims.argmax(dim=1)
        sums = torch.zeros((n_clusters, d), dtype=torch.float64, device=e.device)
        counts = torch.zeros(n_clusters, dtype=torch.float64, device=e.device)
        sums.index_add_(0, labels, e.to(torch.float64))
        counts.index_add_(0, labels, torch.ones_like(labels, dtype=torch.float64))
        if dist is not None:  # the RCCL all-reduce (dedup_actor.py:197-232)
            dist.all_reduce(sums, group=process_group)
            dist.all_reduce(counts, group=process_group)
        nz = counts > 0
        new = (sums[nz] / counts[nz, None]).to(e.dtype)
        cent[nz] = torch.nn.functional.normalize(new, dim=1, eps=1e-12)
    return cent, labels


def pairwise_max_earlier(
    e_sorted_dev: torch.Tensor,
) -> tuple[torch.Tensor, torch.Tensor]:
    """HIP strict-upper-triangular max-cosine scan (cc_pairwise_max_earlier).

    e_sorted_dev: (m, d) f32 CUDA tensor in scan order (L2-normalization is
    applied here, matching dedup_actor.py:399-407).  Returns
    (maxv f32 [m], argi i32 [m]) on device.
    """
    lib = hotpath.require_gpu()
    e = torch.nn.functional.normalize(e_sorted_dev.float(), dim=1, eps=1e-12).contiguous()
    m, d = e.shape
    maxv = torch.empty(m, dtype=torch.float32, device=e.device)
    argi = torch.empty(m, dtype=torch.int32, device=e.device)
    stream = torch.cuda.current_stream(e.device).cuda_stream
    hotpath.check(
        lib.cc_pairwise_max_earlier(e.data_ptr(), m, d, maxv.data_ptr(), argi.data_ptr(), stream)
    )
    return maxv, argi


def dedup_cluster(
    emb_sorted: torch.Tensor, eps: float
) -> tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """One cluster's pruning pass (dedup_actor.py:315-460 semantics).

    Returns (kept_mask bool [m], maxv, argi).  Row kept iff
    max-cosine <= 1 - eps.
    """
    maxv, argi = pairwise_max_earlier(emb_sorted)
    kept = maxv <= (1.0 - eps)
    return kept, maxv, argi


def combine_keep_masks(keep: torch.Tensor, process_group=None) -> torch.Tensor:
    """AND the per-rank keep masks (each rank pruned only the clusters it
    owns, leaving the rest True) — the dedup_pipeline.py N>1 combine,
    RCCL MIN all-reduce on GPU ranks / gloo in the CPU tests."""
    if process_group is None:
        return keep
    ki = keep.int()
    torch.distributed.all_reduce(ki, op=torch.distributed.ReduceOp.MIN,
                                 group=process_group)
    return ki.bool()


def semdedup(
    embeddings: torch.Tensor,
    config: SemDedupConfig,
    process_group=None,
) -> dict:
    """Full pipeline for one rank's shard: k-means -> per-cluster pruning.

    Clusters are processed round-robin by rank (dedup_pipeline.py:237-300
    assigns cluster ranges to actors).  Returns counts + per-row keep mask
    aligned with the input order.
    """
    cent, labels = kmeans_fit(
        embeddings, config.n_clusters, config.n_iters, config.random_seed,
        process_group,
    )
    e = torch.nn.functional.normalize(embeddings.float(), dim=1, eps=1e-12)
    cos_to_cent = (e * cent[labels]).sum(dim=1)
    dist_to_cent = 1.0 - cos_to_cent

    keep = torch.ones(len(e), dtype=torch.bool, device=e.device)
    rank = torch.distributed.get_rank(process_group) if process_group else 0
    world = torch.distributed.get_world_size(process_group) if process_group else 1
    kept_n = 0
    total_n = 0
    for cid in range(rank, config.n_clusters, world):
        sel = (labels == cid).nonzero(as_tuple=True)[0]
        if len(sel) == 0:
            continue
        order = torch.argsort(dist_to_cent[sel], descending=True, stable=True)
        rows = sel[order]
        kept_mask, _, _ = dedup_cluster(e[rows], config.eps)
        keep[rows] = kept_mask
        kept_n += int(kept_mask.sum().item())
        total_n += len(rows)
    return {"kept": kept_n, "total": total_n, "keep_mask": keep, "labels": labels}
