"""BASELINE config #5 scale: semantic dedup over 1M clip embeddings on
one MI355X (k-means + per-cluster pairwise max-cosine via the f32-MFMA
kernel).  Prints one JSON line; the multi-GPU path shards clusters
round-robin over ranks with the k-means all-reduce on RCCL
(tests/test_semdedup_gloo.py covers world_size 2 on CPU)."""
import json
import pathlib
import sys
import time

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
import torch  # noqa: E402

from cosmos_curate_amd.pipelines.video.dedup.semdedup import (  # noqa: E402
    SemDedupConfig,
    semdedup,
)


def main(m: int = 1_000_000, d: int = 512, n_clusters: int = 1000) -> None:
    torch.manual_seed(11)
    # synthetic corpus with planted duplicates (10% near-dupes)
    base = torch.randn(m, d, device="cuda")
    dup_src = torch.randint(0, m, (m // 10,), device="cuda")
    base[: m // 10] = base[dup_src] + 0.01 * torch.randn(m // 10, d, device="cuda")
    cfg = SemDedupConfig(n_clusters=n_clusters, n_iters=10, eps=0.05)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    out = semdedup(base, cfg)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(json.dumps({
        "workload": f"semdedup {m}x{d} f32, {n_clusters} clusters (config #5 scale, 1 GPU)",
        "seconds": round(dt, 2),
        "embeddings_per_s": round(m / dt),
        "kept": int(out["kept"]), "total": int(out["total"]),
        "removed_frac": round(1 - out["kept"] / out["total"], 4),
    }))


if __name__ == "__main__":
    main()
