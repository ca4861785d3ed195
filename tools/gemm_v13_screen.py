"""V13/V14 register-staged GEMM: refcheck + determinism screen + perf A/B
vs the production t256 kernel at the batch-64 bench shapes.

Sync structure is NEW (two-lane discipline): correctness = refcheck vs
production at small/odd sizes, then multi-run determinism at the big
ViT grids (the screen that caught v10's counted-vmcnt races,
profiles/r01_t256_det2.log).

    gpurun -- 'python tools/gemm_v13_screen.py [--perf-only]'
"""

from __future__ import annotations

import ctypes
import pathlib
import subprocess
import sys

import torch

sys.path.insert(0, ".")

from cosmos_curate_amd import hotpath  # noqa: E402

ROOT = pathlib.Path(__file__).resolve().parent.parent
SO = ROOT / "tools" / "libgemm_v13.so"

B64 = [
    ("patch64", 1344 * 49, 768, 3072),
    ("qkv64", 1344 * 50, 2304, 768),
    ("out64", 1344 * 50, 768, 768),
    ("fc1_64", 1344 * 50, 3072, 768),
    ("fc2_64", 1344 * 50, 768, 3072),
    ("square4k", 4096, 4096, 4096),
    ("square8k", 8192, 8192, 8192),
]


def build() -> ctypes.CDLL:
    src = ROOT / "tools" / "gemm_v13.hip"
    if not SO.exists() or SO.stat().st_mtime < src.stat().st_mtime:
        subprocess.run(
            ["hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
             f"-I{ROOT}", "-shared", str(src), "-o", str(SO)],
            check=True,
        )
    lib = ctypes.CDLL(str(SO))
    lib.cc_gemm_v13.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
        ctypes.c_long, ctypes.c_long, ctypes.c_long,
        ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_uint64,
    ]
    lib.cc_gemm_v13.restype = ctypes.c_int
    lib.cc_gemm_v15.argtypes = lib.cc_gemm_v13.argtypes
    lib.cc_gemm_v15.restype = ctypes.c_int
    lib.cc_gemm_v16.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
        ctypes.c_long, ctypes.c_long, ctypes.c_long,
        ctypes.c_int, ctypes.c_int, ctypes.c_uint64,
    ]
    lib.cc_gemm_v16.restype = ctypes.c_int
    lib.cc_gemm_v17.argtypes = lib.cc_gemm_v13.argtypes
    lib.cc_gemm_v17.restype = ctypes.c_int
    lib.cc_gemm_v19.argtypes = lib.cc_gemm_v13.argtypes
    lib.cc_gemm_v19.restype = ctypes.c_int
    lib.cc_gemm_v20.argtypes = lib.cc_gemm_v13.argtypes
    lib.cc_gemm_v20.restype = ctypes.c_int
    lib.cc_gemm_v21.argtypes = lib.cc_gemm_v13.argtypes
    lib.cc_gemm_v21.restype = ctypes.c_int
    return lib


def refcheck(prod, v13, stream) -> int:
    shapes = [(256, 256, 256), (512, 512, 512), (512, 256, 768),
              (1000, 777, 768), (260, 300, 384), (4096, 4096, 4096),
              (300, 300, 128)]
    bad = 0
    for persist in (0, 1):
        for (M, N, K) in shapes:
            torch.manual_seed(M + persist)
            a = (torch.randn(M, K) * 0.3).to(torch.bfloat16).cuda()
            b = (torch.randn(N, K) * 0.3).to(torch.bfloat16).cuda()
            c1 = torch.empty((M, N), dtype=torch.bfloat16, device="cuda")
            c2 = torch.full((M, N), 7.0, dtype=torch.bfloat16, device="cuda")
            hotpath.check(prod.cc_gemm_bf16(a.data_ptr(), b.data_ptr(),
                                            c1.data_ptr(), M, N, K, None, 1,
                                            stream))
            for name, fn in (("v13", v13.cc_gemm_v13), ("v15", v13.cc_gemm_v15),
                             ("v17", v13.cc_gemm_v17),
                             ("v19", v13.cc_gemm_v19),
                             ("v20", v13.cc_gemm_v20),
                             ("v21", v13.cc_gemm_v21), ("v16", None)):
                c2.fill_(7.0)
                if name == "v16":
                    rc = v13.cc_gemm_v16(a.data_ptr(), b.data_ptr(),
                                         c2.data_ptr(), M, N, K, 1, persist,
                                         stream)
                else:
                    rc = fn(a.data_ptr(), b.data_ptr(), c2.data_ptr(),
                            M, N, K, 1, persist, 0, stream)
                if rc == -2:
                    continue  # v15 needs K%128==0, KT>=4
                assert rc == 0, (name, M, N, K, rc)
                torch.cuda.synchronize()
                err = (c1.float() - c2.float()).abs().max().item()
                if err > 1e-3:
                    bad += 1
                    print(f"REFCHECK FAIL {name} p{persist} {M}x{N}x{K}: max {err}")
    print("refcheck:", "FAIL" if bad else "PASS")
    return bad


def determinism(v13, stream) -> int:
    shapes = [(65856, 768, 3072), (67200, 2304, 768), (67200, 3072, 768)]
    bad = 0
    for persist in (0, 1):
        for remap in (0, 1, 2):
            for (M, N, K) in shapes:
                torch.manual_seed(42)
                a = (torch.randn(M, K) * 0.3).to(torch.bfloat16).cuda()
                b = (torch.randn(N, K) * 0.3).to(torch.bfloat16).cuda()
                for name, fn in (("v13", v13.cc_gemm_v13),
                                 ("v15", v13.cc_gemm_v15),
                                 ("v17", v13.cc_gemm_v17),
                                 ("v19", v13.cc_gemm_v19),
                                 ("v20", v13.cc_gemm_v20),
                                 ("v21", v13.cc_gemm_v21),
                                 ("v16", None)):
                    if name == "v16" and remap > 1:
                        continue
                    ref = None
                    for run in range(6):
                        c = torch.full((M, N), 9.0, dtype=torch.bfloat16,
                                       device="cuda")
                        if name == "v16":
                            rc = v13.cc_gemm_v16(a.data_ptr(), b.data_ptr(),
                                                 c.data_ptr(), M, N, K, 1,
                                                 remap, stream)
                        else:
                            rc = fn(a.data_ptr(), b.data_ptr(),
                                    c.data_ptr(), M, N, K, 1, persist,
                                    remap, stream)
                        assert rc == 0
                        torch.cuda.synchronize()
                        if ref is None:
                            ref = c
                        else:
                            diff = (ref != c).sum().item()
                            if diff:
                                bad += 1
                                print(f"NONDET {name} p{persist} r{remap} "
                                      f"{M}x{N}x{K} run{run}: {diff} differ")
                                break
    print("determinism:", "FAIL" if bad else "PASS")
    return bad


def perf(prod, v13, stream) -> None:
    ev0, ev1 = torch.cuda.Event(True), torch.cuda.Event(True)
    print(f"{'shape':9s} {'M':>6s} {'N':>5s} {'K':>5s} "
          f"{'prod':>7s} {'v13':>7s} {'v13p':>7s} {'v13pr':>7s} {'v15p':>7s} {'v15pr':>7s} {'v15pc':>7s} {'v16g':>7s} {'v16x':>7s} {'v17p':>7s} {'v17pr':>7s} {'v19p':>7s} {'v19pr':>7s} {'v20p':>7s} {'v20pr':>7s} {'v21p':>7s} {'v21pr':>7s}")
    for label, M, N, K in B64:
        torch.manual_seed(1)
        a = torch.randn(M, K).to(torch.bfloat16).cuda()
        b = torch.randn(N, K).to(torch.bfloat16).cuda()
        c = torch.empty((M, N), dtype=torch.bfloat16, device="cuda")
        flops = 2.0 * M * N * K
        iters = 30 if M * N * K < 2 ** 36 else 10

        def time_one(fn):
            for _ in range(5):
                fn()
            torch.cuda.synchronize()
            ev0.record()
            for _ in range(iters):
                fn()
            ev1.record()
            torch.cuda.synchronize()
            return flops * iters / (ev0.elapsed_time(ev1) / 1e3) / 1e12

        tf_prod = time_one(lambda: hotpath.check(prod.cc_gemm_bf16(
            a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, None, 1,
            stream)))
        tf_13 = time_one(lambda: v13.cc_gemm_v13(
            a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, 1, 0, 0,
            stream))
        tf_13p = time_one(lambda: v13.cc_gemm_v13(
            a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, 1, 1, 0,
            stream))
        tf_13pr = time_one(lambda: v13.cc_gemm_v13(
            a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, 1, 1, 1,
            stream))
        tf_15p = time_one(lambda: v13.cc_gemm_v15(
            a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, 1, 1, 0,
            stream))
        tf_15pr = time_one(lambda: v13.cc_gemm_v15(
            a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, 1, 1, 1,
            stream))
        tf_15pc = time_one(lambda: v13.cc_gemm_v15(
            a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, 1, 1, 2,
            stream))
        tf_16g = time_one(lambda: v13.cc_gemm_v16(
            a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, 1, 0,
            stream))
        tf_16x = time_one(lambda: v13.cc_gemm_v16(
            a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, 1, 1,
            stream))
        tf_17p = time_one(lambda: v13.cc_gemm_v17(
            a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, 1, 1, 0,
            stream))
        tf_17pr = time_one(lambda: v13.cc_gemm_v17(
            a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, 1, 1, 1,
            stream))
        tf_19p = time_one(lambda: v13.cc_gemm_v19(
            a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, 1, 1, 0,
            stream))
        tf_19pr = time_one(lambda: v13.cc_gemm_v19(
            a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, 1, 1, 1,
            stream))
        tf_20p = time_one(lambda: v13.cc_gemm_v20(
            a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, 1, 1, 0,
            stream))
        tf_20pr = time_one(lambda: v13.cc_gemm_v20(
            a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, 1, 1, 1,
            stream))
        tf_21p = time_one(lambda: v13.cc_gemm_v21(
            a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, 1, 1, 0,
            stream))
        tf_21pr = time_one(lambda: v13.cc_gemm_v21(
            a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, 1, 1, 1,
            stream))
        print(f"{label:9s} {M:6d} {N:5d} {K:5d} {tf_prod:7.0f} {tf_13:7.0f} "
              f"{tf_13p:7.0f} {tf_13pr:7.0f} {tf_15p:7.0f} {tf_15pr:7.0f} "
              f"{tf_15pc:7.0f} {tf_16g:7.0f} {tf_16x:7.0f} {tf_17p:7.0f} "
              f"{tf_17pr:7.0f} {tf_19p:7.0f} {tf_19pr:7.0f} {tf_20p:7.0f} "
              f"{tf_20pr:7.0f} {tf_21p:7.0f} {tf_21pr:7.0f}")


def main() -> None:
    prod = hotpath.require_gpu()
    v13 = build()
    stream = torch.cuda.current_stream().cuda_stream
    if "--perf-only" not in sys.argv:
        if refcheck(prod, v13, stream):
            return
        if determinism(v13, stream):
            return
    perf(prod, v13, stream)


if __name__ == "__main__":
    main()
