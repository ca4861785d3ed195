"""A/B CC_GEMM_PERSIST=1 vs 0 with bench-faithful GEMM calls: full
epilogues (bias/act/residual), bf16 out, fresh activation-like operands,
plus a whole ViT-B/32 tower forward timing.  Run as parent (spawns two
children with the env set) or child (--child)."""
import ctypes
import os
import subprocess
import sys

sys.path.insert(0, ".")


def child() -> None:
    import torch

    from cosmos_curate_amd import hotpath

    lib = hotpath.require_gpu()
    stream = torch.cuda.current_stream().cuda_stream
    N_FRAMES = 1344
    SHAPES = [
        # label, M, N, K, act, has_bias, has_res
        ("patch", N_FRAMES * 49, 768, 3072, 0, 1, 0),
        ("qkv", N_FRAMES * 50, 2304, 768, 0, 1, 0),
        ("attn_out", N_FRAMES * 50, 768, 768, 0, 1, 1),
        ("fc1", N_FRAMES * 50, 3072, 768, 1, 1, 0),
        ("fc2", N_FRAMES * 50, 768, 3072, 0, 1, 1),
    ]
    print(f"CC_GEMM_PERSIST={os.environ.get('CC_GEMM_PERSIST', '<unset>')}")
    for label, M, N, K, act, hb, hr in SHAPES:
        torch.manual_seed(2)
        a = (torch.randn(M, K) * 0.1).to(torch.bfloat16).cuda()
        b = (torch.randn(N, K) * 0.1).to(torch.bfloat16).cuda()
        bias = torch.randn(N).float().cuda() if hb else None
        res = (torch.randn(M, N) * 0.1).to(torch.bfloat16).cuda() if hr else None
        c = torch.empty((M, N), dtype=torch.bfloat16, device="cuda")
        call = lambda: hotpath.check(lib.cc_gemm_bf16_ex(
            a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K,
            bias.data_ptr() if hb else None, 1, act,
            res.data_ptr() if hr else None, stream))
        for _ in range(5):
            call()
        torch.cuda.synchronize()
        hotpath.timing_enable(True)
        for _ in range(20):
            call()
        torch.cuda.synchronize()
        ms, cnt = hotpath.timing_report("gemm_bf16")
        hotpath.timing_enable(False)
        tf = 2.0 * M * N * K * cnt / (ms / 1e3) / 1e12
        print(f"  {label:9s} {tf:7.0f} TF ({ms/cnt*1000:6.0f} us/launch)")

    # whole tower forward (the bench's GEMM context)
    from cosmos_curate_amd.models.clip_vit import ClipVisionTowerAMD
    from cosmos_curate_amd.models.clip_weights import (
        CONFIGS,
        make_clip_vit_b32_weights,
    )

    tower = ClipVisionTowerAMD(make_clip_vit_b32_weights(),
                               CONFIGS["vit_b32"]).to("cuda")
    torch.manual_seed(3)
    pixels = (torch.randn(N_FRAMES, 3, 224, 224) * 0.5).to(torch.bfloat16).cuda()
    with torch.no_grad():
        for _ in range(3):
            tower(pixels)
        torch.cuda.synchronize()
        hotpath.timing_enable(True)
        ev0, ev1 = torch.cuda.Event(True), torch.cuda.Event(True)
        ev0.record()
        for _ in range(5):
            tower(pixels)
        ev1.record()
        torch.cuda.synchronize()
    ms, cnt = hotpath.timing_report("gemm_bf16")
    hotpath.timing_enable(False)
    gemm_flop_per_fwd = 8.727e9 / 21 * 1344 * 21 / 1344  # per frame * frames
    total_gemm_flop = 8.727e9 * N_FRAMES * 5  # ~8.73 GF/frame
    tf = total_gemm_flop / (ms / 1e3) / 1e12
    print(f"  tower fwd: {ev0.elapsed_time(ev1)/5:.2f} ms; gemm "
          f"{ms/5:.2f} ms over {cnt/5:.0f} launches -> {tf:.0f} TF "
          f"(frac {tf/2500:.3f})")


def main() -> None:
    if "--child" in sys.argv:
        child()
        return
    for mode in ("0", "1"):
        env = dict(os.environ, CC_GEMM_PERSIST=mode)
        subprocess.run([sys.executable, __file__, "--child"], env=env,
                       check=True)


if __name__ == "__main__":
    main()
