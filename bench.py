"""Flagship benchmark: the split/embed hot path on MI355X.

Measures BASELINE.json's metric ("clips/sec + embedded frames/sec, 1080p30
H.264") on the workload of BASELINE configs[1] — the largest single-GPU
configuration: per clip, 21 sampled 1080p NV12 frames (10 s @ 30 fps
sampled at 2 fps with the endpoint rule) -> fused NV12->RGB + bilinear
resize (HIP) -> fused normalize + patch-extraction straight into the
GEMM A layout (HIP) -> ViT forward entirely on hand-written kernels
(MFMA bf16 GEMMs with fused epilogues, fused LayerNorm, LDS-resident /
streaming attention by sequence length) -> mean-pooled L2-normalized
clip embedding (copied to host).  --model selects the tower: vit_b32
(flagship), vit_l14, siglip_l16_256.

The timed region starts with NV12 surfaces already resident in HBM (tier
contract; H.264 decode itself needs librocdecode, absent from this image —
see DESIGN.md "decode roofline" for how the seam is accounted).  A step =
one batch of --clips clips.  Multi-GPU = weak scaling: each rank processes
its own clips; no data-path collective.

Contract: one JSON line from rank 0 with metric/value/unit/... plus
`roofline` (dominant kernel = gemm_bf16, HIP-event timed on its launch
stream inside the timed region) and `cpu_baseline` (oracle timed on host
cores, rank 0, N=1 only).
"""

from __future__ import annotations

import argparse
import json
import os
import time

import numpy as np
import torch

FRAMES_PER_CLIP = 21  # 10 s @ 30 fps sampled at 2 fps, endpoint included
SRC_H, SRC_W = 1088, 1920  # H.264 coded size for 1080p
RES = 224


def gemm_flops_per_frame(variant: str = "vit_b32") -> float:
    """cc_gemm FLOPs (2*M*N*K) per frame for the tower geometry."""
    from cosmos_curate_amd.models import clip_weights as cw

    cfg = cw.CONFIGS[variant]
    g = cfg.image // cfg.patch
    patches, tokens = g * g, cfg.num_pos  # num_pos handles no-CLS towers
    # ALGORITHMIC K for the patch embed (3*patch^2), NOT the kernel's
    # 64-multiple padding: padded columns multiply zeros and must not
    # inflate `achieved` (tier contract: algorithmic flops only).
    patch_k = 3 * cfg.patch * cfg.patch
    per_layer = (
        2 * tokens * cfg.hidden * (3 * cfg.hidden)  # fused qkv
        + 2 * tokens * cfg.hidden * cfg.hidden      # out proj
        + 2 * tokens * cfg.hidden * cfg.intermediate
        + 2 * tokens * cfg.intermediate * cfg.hidden
    )
    return float(
        2 * patches * patch_k * cfg.hidden + cfg.layers * per_layer
        + 2 * cfg.hidden * cfg.proj
    )


def make_nv12_batch(n_frames: int, seed: int) -> tuple[np.ndarray, np.ndarray]:
    """Seeded moving-gradient+noise NV12 frames (BASELINE.md corpus recipe)."""
    rng = np.random.default_rng(seed)
    yy, xx = np.mgrid[0:SRC_H, 0:SRC_W]
    base = ((xx * 255 // SRC_W + yy // 3) % 256).astype(np.uint8)
    t = (np.arange(n_frames, dtype=np.int32) * 5)[:, None, None]
    y = ((base[None].astype(np.int32) + t) % 256).astype(np.uint8)
    noise = rng.integers(-10, 11, size=(n_frames, SRC_H // 8, SRC_W // 8), dtype=np.int16)
    y = np.clip(
        y.astype(np.int16) + np.kron(noise, np.ones((8, 8), dtype=np.int16)), 0, 255
    ).astype(np.uint8)
    uv = rng.integers(80, 176, size=(n_frames, SRC_H // 2, SRC_W), dtype=np.uint8).astype(np.uint8)
    return y, uv


def run_cpu_baseline(n_clips: int = 8) -> dict:
    """Oracle (CPU restatement) on the same per-clip work, host cores."""
    from oracle import color as ocolor
    from oracle import vit as oracle_vit
    from cosmos_curate_amd.models.clip_weights import make_clip_vit_b32_weights

    ref = oracle_vit.build_reference_clip_vision(make_clip_vit_b32_weights())
    y, uv = make_nv12_batch(FRAMES_PER_CLIP, seed=0)
    res = 224  # the baseline is always the flagship B/32 oracle
    t0 = time.perf_counter()
    for _ in range(n_clips):
        rgb = np.stack(
            [
                ocolor.resize_bilinear_u8(
                    ocolor.nv12_to_rgb(y[j], uv[j].reshape(SRC_H // 2, SRC_W // 2, 2)),
                    res, res,
                )
                for j in range(FRAMES_PER_CLIP)
            ]
        )
        pix = ocolor.clip_preprocess(rgb)
        emb = oracle_vit.embed_frames_fp32(ref, pix)
        pooled = emb.mean(axis=0)
        pooled /= np.linalg.norm(pooled)
    dt = time.perf_counter() - t0
    return {
        "value": round(n_clips / dt, 4),
        "unit": "clips/s",
        "cores": torch.get_num_threads(),
        "kind": "port",
        "sample": f"{n_clips} clips x {FRAMES_PER_CLIP} 1080p NV12 frames, "
        f"numpy pixel oracle + transformers fp32 ViT-B/32 ({dt:.1f}s)",
    }


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument(
        "--clips", type=int, default=224,
        help="clips per step per rank (default 224: the persistent GEMM "
        "amortizes per-tile costs over more tiles per WG; clips/s "
        "plateaus from batch 160 and the GEMM roofline fraction peaks "
        "around 224 — profiles/r02_batch_sweep*.log, r02_batch3.log)",
    )
    ap.add_argument(
        "--host-fed", action="store_true",
        help="measurement leg: NV12 starts in PINNED HOST memory and the "
        "PCIe upload runs inside the timed region (boundary-hands-over-"
        "host-buffers rate; reported as its own line, never the headline "
        "value — tier contract / DESIGN.md §4)",
    )
    ap.add_argument("--skip-cpu-baseline", action="store_true")
    ap.add_argument(
        "--graphs", action="store_true", default=True,
        help="capture the per-step kernel sequence in a hipGraph and replay it "
        "(launch-bound inner loop -> graph, per the MI355X design brief); "
        "the roofline leg still runs eager with HIP-event timing",
    )
    ap.add_argument("--no-graphs", dest="graphs", action="store_false")
    ap.add_argument(
        "--streams", type=int, default=1, choices=[1, 2],
        help="split each step's batch across N HIP streams (overlaps one "
        "half's GEMM chain with the other's small kernels)",
    )
    ap.add_argument(
        "--model", default="vit_b32",
        choices=["vit_b32", "vit_l14", "siglip_l16_256"],
        help="embedder tower: vit_b32 (flagship, configs #1/#2), vit_l14 "
        "(the reference's CLIP model) or siglip_l16_256 (config #3's "
        "SigLIP-L class)",
    )
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    dist = None
    if world > 1:
        import torch.distributed as dist_mod

        dist = dist_mod
        dist.init_process_group("nccl")
    torch.cuda.set_device(local_rank)
    device = torch.device("cuda", local_rank)

    from cosmos_curate_amd import hotpath
    from cosmos_curate_amd.models.clip import _CLIPImageEmbeddings

    lib = hotpath.require_gpu()
    model = _CLIPImageEmbeddings(args.model)
    global RES
    RES = model.tower.cfg.image  # 224 (CLIP) / 256 (SigLIP)

    B = args.clips
    F = B * FRAMES_PER_CLIP
    # inputs resident in HBM before the timed region (tier contract);
    # --host-fed instead stages them in pinned host memory and uploads
    # inside the timed region (PCIe-inclusive leg)
    y_host, uv_host = make_nv12_batch(FRAMES_PER_CLIP, seed=1000 + rank)
    if args.host_fed:
        args.graphs = False  # H2D copies stay eager in this leg
        y_pin = torch.from_numpy(y_host).repeat(B, 1, 1).contiguous().pin_memory()
        uv_pin = torch.from_numpy(uv_host).repeat(B, 1, 1).contiguous().pin_memory()
        y_dev = torch.empty_like(y_pin, device=device)
        uv_dev = torch.empty_like(uv_pin, device=device)
    else:
        y_dev = torch.from_numpy(y_host).to(device).repeat(B, 1, 1).contiguous()
        uv_dev = torch.from_numpy(uv_host).to(device).repeat(B, 1, 1).contiguous()
    rgb = torch.empty((F, RES, RES, 3), dtype=torch.uint8, device=device)
    stream = torch.cuda.current_stream(device).cuda_stream

    side_streams = [torch.cuda.Stream(device) for _ in range(args.streams - 1)]

    def step_device() -> torch.Tensor:
        if args.host_fed:
            y_dev.copy_(y_pin, non_blocking=True)
            uv_dev.copy_(uv_pin, non_blocking=True)
        hotpath.check(
            lib.cc_nv12_to_rgb_resize(
                y_dev.data_ptr(), uv_dev.data_ptr(), F, SRC_H, SRC_W, SRC_W,
                rgb.data_ptr(), RES, RES, stream,
            )
        )
        if args.streams == 1:
            # fused normalize+patch-extract straight into the GEMM layout
            patches = model.preprocess_patches_u8(rgb)
            emb = model.tower(patches=patches, n=F)  # (F, proj) f32 unit-norm
        else:
            # split the batch across streams: one half's bandwidth-bound
            # kernels overlap the other half's MFMA chain
            main = torch.cuda.current_stream(device)
            half = F // 2
            parts = []
            chunks = [rgb[:half], rgb[half:]]
            for st, chunk in zip([main, *side_streams], chunks):
                with torch.cuda.stream(st):
                    if st is not main:
                        st.wait_stream(main)
                    parts.append(model.tower(
                        patches=model.preprocess_patches_u8(chunk),
                        n=chunk.shape[0]))
            for st in side_streams:
                main.wait_stream(st)
            emb = torch.cat(parts, dim=0)
        per_clip = emb.view(B, FRAMES_PER_CLIP, emb.shape[-1]).mean(dim=1)
        return per_clip / torch.linalg.vector_norm(per_clip, dim=-1, keepdim=True)

    def step() -> np.ndarray:
        return step_device().cpu().numpy()  # embeddings leave the device (16 KB)

    ref_out = step()  # one mandatory warm/reference step
    for _ in range(max(0, args.warmup - 1)):
        step()

    if args.graphs:
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            static_out = step_device()

        def step() -> np.ndarray:  # noqa: F811 — graph-replay step
            graph.replay()
            return static_out.cpu().numpy()

        # sanity: replay reproduces the eager output (same inputs every step)
        got = step()
        cos = float((got * ref_out).sum(axis=1).min())
        assert cos > 0.9999, f"graph replay diverged from eager: {cos}"

    torch.cuda.synchronize(device)
    if dist:
        dist.barrier()

    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    torch.cuda.synchronize(device)
    if dist:
        dist.barrier()
    elapsed = time.perf_counter() - t0
    if dist:
        t = torch.tensor([elapsed], device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    # roofline leg: HIP-event per-launch timing of the dominant kernel, on
    # its launch stream (eager; graphs cannot carry per-launch events)
    hotpath.timing_enable(True)
    roofline_steps = max(2, args.steps // 5)
    for _ in range(roofline_steps):
        step_device().cpu()
    torch.cuda.synchronize(device)
    hotpath.timing_enable(False)

    gemm_ms, gemm_count = hotpath.timing_report("gemm_bf16")
    total_clips = B * args.steps * world
    clips_per_s = total_clips / elapsed
    frames_per_s = clips_per_s * FRAMES_PER_CLIP

    gemm_flops_step = gemm_flops_per_frame(args.model) * F
    gemm_flops_total = gemm_flops_step * roofline_steps  # instrumented phase
    achieved = gemm_flops_total / (gemm_ms / 1e3) if gemm_ms > 0 else 0.0
    peak = 2.5e15  # dense bf16 MFMA peak, MI355X_MICROARCH.md (spec; 2495 TF measured)

    if rank == 0:
        cpu_baseline = (
            None
            if (args.skip_cpu_baseline or world > 1)
            else run_cpu_baseline()
        )
        out = {
            "metric": "clips/sec + embedded frames/sec, 1080p30 H.264, 1/2/4/8 MI355X",
            "value": round(clips_per_s, 3),
            "unit": "clips/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference publishes no number
            "dtype": "bf16",
            "data": "synthetic",
            "leg": "host_fed" if args.host_fed else "hbm_resident",
            "config": {
                "workload": f"split_pipeline {world}xMI355X: HIP NV12->RGB/resize + "
                f"{ {'vit_b32': 'CLIP-ViT-B/32', 'vit_l14': 'CLIP-ViT-L/14', 'siglip_l16_256': 'SigLIP-L/16-256'}[args.model] } MFMA bf16 "
                "(decode seam excluded: no librocdecode in image)",
                "model": args.model,
                "clips_per_step": B,
                "frames_per_clip": FRAMES_PER_CLIP,
                "src": ("1080p30 NV12 pinned-host (PCIe upload in timed "
                        "region)" if args.host_fed else "1080p30 NV12 in HBM"),
                "resolution": RES,
                "parallelism": f"dp{world}",
                "embedded_frames_per_s": round(frames_per_s, 1),
            },
            "roofline": {
                "bound": "mfma",
                "achieved": round(achieved, 3),
                "peak": peak,
                "unit": "FLOP/s",
                "frac": round(achieved / peak, 4),
                "traffic": None,
                "kernel": "gemm_bf16",
                "launches": int(gemm_count),
                "kernel_ms_total": round(gemm_ms, 3),
            },
            "cpu_baseline": cpu_baseline,
        }
        print(json.dumps(out))
    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
