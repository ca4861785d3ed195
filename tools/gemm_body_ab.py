"""A/B the two GEMM bodies (CC_GEMM_WIDE) per ViT shape incl. epilogues."""
import ctypes
import os
import sys
import time

import torch

sys.path.insert(0, ".")
from cosmos_curate_amd import hotpath  # noqa: E402

SHAPES = [  # (label, M, N, K, act, residual?)
    ("patch", 64 * 21 * 49, 768, 3072, 0, False),
    ("qkv", 64 * 21 * 50, 2304, 768, 0, False),
    ("attn_out+res", 64 * 21 * 50, 768, 768, 0, True),
    ("fc1+gelu", 64 * 21 * 50, 3072, 768, 1, False),
    ("fc2+res", 64 * 21 * 50, 768, 3072, 0, True),
]


def main():
    lib = hotpath.require_gpu()
    stream = torch.cuda.current_stream().cuda_stream
    print(f"{'shape':14s} {'16x16':>8s} {'32x32':>8s}  TF/s (best of 3)")
    for label, M, N, K, act, use_res in SHAPES:
        torch.manual_seed(1)
        a = (torch.randn(M, K) * 0.3).to(torch.bfloat16).cuda()
        b = (torch.randn(N, K) * 0.3).to(torch.bfloat16).cuda()
        bias = torch.randn(N).float().cuda()
        res = torch.randn(M, N).to(torch.bfloat16).cuda() if use_res else None
        c = torch.empty((M, N), dtype=torch.bfloat16, device="cuda")

        def call():
            rc = lib.cc_gemm_bf16_ex(
                a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K,
                bias.data_ptr(), 1, act,
                res.data_ptr() if res is not None else None, stream)
            assert rc == 0

        results = []
        for wide in ("0", "1"):
            os.environ["CC_GEMM_WIDE"] = wide
            # env read is cached per-process after first launch; re-exec trick:
            # instead call through a fresh value by... the static caches.
            # So run both bodies via subprocess if cached — detect below.
            for _ in range(5):
                call()
            torch.cuda.synchronize()
            best = 1e9
            for _ in range(3):
                t0 = time.perf_counter()
                for _ in range(30):
                    call()
                torch.cuda.synchronize()
                best = min(best, (time.perf_counter() - t0) / 30)
            results.append(2.0 * M * N * K / best / 1e12)
        print(f"{label:14s} {results[0]:8.1f} {results[1]:8.1f}")


if __name__ == "__main__":
    main()
