"""Compare GEMM structure variants (tools/gemm_variants.hip) vs the
production kernel on the ViT shapes.  Builds the variants .so on the fly.

    gpurun -- 'python tools/gemm_bench2.py'
"""

from __future__ import annotations

import ctypes
import pathlib
import subprocess
import sys
import time

import torch

sys.path.insert(0, ".")

from cosmos_curate_amd import hotpath  # noqa: E402

ROOT = pathlib.Path(__file__).resolve().parent.parent
SO = ROOT / "tools" / "libgemm_variants.so"
SO8 = ROOT / "tools" / "libgemm_v8.so"

SHAPES = [
    ("patch", 336 * 49, 768, 3072),
    ("qkv", 336 * 50, 2304, 768),
    ("attn_out", 336 * 50, 768, 768),
    ("fc1", 336 * 50, 3072, 768),
    ("fc2", 336 * 50, 768, 3072),
    ("square4k", 4096, 4096, 4096),
    ("square8k", 8192, 8192, 8192),
    ("patch64", 1344 * 49, 768, 3072),
    ("qkv64", 1344 * 50, 2304, 768),
    ("out64", 1344 * 50, 768, 768),
    ("fc1_64", 1344 * 50, 3072, 768),
    ("fc2_64", 1344 * 50, 768, 3072),
]


def build() -> ctypes.CDLL:
    src = ROOT / "tools" / "gemm_variants.hip"
    if not SO.exists() or SO.stat().st_mtime < src.stat().st_mtime:
        subprocess.run(
            ["hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
             f"-I{ROOT}", "-shared", str(src), "-o", str(SO)],
            check=True,
        )
    lib = ctypes.CDLL(str(SO))
    lib.cc_gemm_variant.argtypes = [
        ctypes.c_int, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
        ctypes.c_int64, ctypes.c_int64, ctypes.c_int64, ctypes.c_int, ctypes.c_uint64,
    ]
    src8 = ROOT / "tools" / "gemm_v8.hip"
    if not SO8.exists() or SO8.stat().st_mtime < src8.stat().st_mtime:
        subprocess.run(
            ["hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
             f"-I{ROOT}", "-shared", str(src8), "-o", str(SO8)],
            check=True,
        )
    lib8 = ctypes.CDLL(str(SO8))
    lib8.cc_gemm_v8.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
        ctypes.c_int64, ctypes.c_int64, ctypes.c_int64, ctypes.c_int, ctypes.c_uint64,
    ]
    src10 = ROOT / "tools" / "gemm_v10.hip"
    SO10 = ROOT / "tools" / "libgemm_v10.so"
    if not SO10.exists() or SO10.stat().st_mtime < src10.stat().st_mtime:
        subprocess.run(
            ["hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
             f"-I{ROOT}", "-shared", str(src10), "-o", str(SO10)],
            check=True,
        )
    lib10 = ctypes.CDLL(str(SO10))
    lib10.cc_gemm_v10.argtypes = lib8.cc_gemm_v8.argtypes
    lib10.cc_gemm_v11.argtypes = lib8.cc_gemm_v8.argtypes
    lib10.cc_gemm_v12.argtypes = lib8.cc_gemm_v8.argtypes
    return lib, lib8, lib10


def time_variant(fn, iters=30) -> float:
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main() -> None:
    prod = hotpath.require_gpu()
    var, var8, var10 = build()
    stream = torch.cuda.current_stream().cuda_stream
    print(f"{'shape':9s} {'M':>6s} {'N':>5s} {'K':>5s} | {'prod':>7s} {'v2':>7s} {'v6':>7s} {'v9':>7s} {'v8':>7s} {'v10':>7s} {'v11':>7s} {'v12':>7s}  TF/s (best of 3 reps)")
    for label, M, N, K in SHAPES:
        torch.manual_seed(1)
        a = (torch.randn(M, K) * 0.3).to(torch.bfloat16).cuda()
        b = (torch.randn(N, K) * 0.3).to(torch.bfloat16).cuda()
        c = torch.empty((M, N), dtype=torch.bfloat16, device="cuda")

        def prod_call():
            hotpath.check(prod.cc_gemm_bf16(a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, None, 1, stream))

        prod_call()
        torch.cuda.synchronize()
        want = c[:128, :128].float().cpu().clone()
        iters = 20 if M * N * K > 2**36 else 40
        results = []
        calls = {0: prod_call}
        for v in (2, 6, 9):
            def vcall(v=v):
                rc = var.cc_gemm_variant(v, a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, 1, stream)
                assert rc == 0
            calls[v] = vcall

        def v8call():
            rc = var8.cc_gemm_v8(a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, 1, stream)
            assert rc == 0
        calls[8] = v8call

        def v10call():
            rc = var10.cc_gemm_v10(a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, 1, stream)
            assert rc == 0
        calls[10] = v10call

        def v11call():
            rc = var10.cc_gemm_v11(a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, 1, stream)
            assert rc == 0
        calls[11] = v11call

        def v12call():
            rc = var10.cc_gemm_v12(a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, 1, stream)
            assert rc == 0
        calls[12] = v12call
        for v, fn in calls.items():
            if v != 0:
                c.zero_()
                fn()
                torch.cuda.synchronize()
                err = (c[:128, :128].float().cpu() - want).abs().max().item()
                if err > 0.2:
                    results.append(float("nan"))
                    print(f"  variant {v} WRONG on {label}: max err {err}")
                    continue
            for _ in range(5):  # warmup per variant
                fn()
            best = min(time_variant(fn, iters) for _ in range(3))
            results.append(2.0 * M * N * K / best / 1e12)
        print(
            f"{label:9s} {M:6d} {N:5d} {K:5d} | "
            + " ".join(f"{r:7.1f}" for r in results)
        )


if __name__ == "__main__":
    main()
