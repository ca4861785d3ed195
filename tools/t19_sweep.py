"""T19 pattern sweep (v22) vs production at the b64 shapes."""
import ctypes
import sys

import torch

sys.path.insert(0, ".")
from cosmos_curate_amd import hotpath  # noqa: E402
from tools.gemm_v13_screen import build  # noqa: E402

SHAPES = [("patch64", 65856, 768, 3072), ("qkv64", 67200, 2304, 768),
          ("fc1_64", 67200, 3072, 768), ("fc2_64", 67200, 768, 3072),
          ("square8k", 8192, 8192, 8192)]


def main():
    prod = hotpath.require_gpu()
    lib = build()
    lib.cc_gemm_v22.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
        ctypes.c_long, ctypes.c_long, ctypes.c_long,
        ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_uint64,
    ]
    lib.cc_gemm_v22.restype = ctypes.c_int
    stream = torch.cuda.current_stream().cuda_stream
    ev0, ev1 = torch.cuda.Event(True), torch.cuda.Event(True)
    print(f"{'shape':9s} {'prod':>6s}" + "".join(f"  pat{p}" for p in range(4)))
    for label, M, N, K in SHAPES:
        torch.manual_seed(1)
        a = (torch.randn(M, K) * 0.3).to(torch.bfloat16).cuda()
        b = (torch.randn(N, K) * 0.3).to(torch.bfloat16).cuda()
        c = torch.empty((M, N), dtype=torch.bfloat16, device="cuda")
        want = None
        flops = 2.0 * M * N * K
        iters = 10 if M * N * K >= 2**39 else 25

        def time_one(fn):
            for _ in range(4):
                fn()
            torch.cuda.synchronize()
            ev0.record()
            for _ in range(iters):
                fn()
            ev1.record()
            torch.cuda.synchronize()
            return flops * iters / (ev0.elapsed_time(ev1) / 1e3) / 1e12

        row = []
        tf = time_one(lambda: hotpath.check(prod.cc_gemm_bf16(
            a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, None, 1, stream)))
        want = c.clone()
        row.append(f"{tf:6.0f}")
        for pat in range(4):
            c.fill_(3.0)
            rc = lib.cc_gemm_v22(a.data_ptr(), b.data_ptr(), c.data_ptr(),
                                 M, N, K, 1, pat, 1, stream)
            assert rc == 0
            torch.cuda.synchronize()
            assert torch.equal(c, want), (label, pat)  # bitwise vs prod
            tf = time_one(lambda: lib.cc_gemm_v22(
                a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, 1, pat, 1,
                stream))
            row.append(f"{tf:6.0f}")
        print(f"{label:9s} " + " ".join(row))


if __name__ == "__main__":
    main()
