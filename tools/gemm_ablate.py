"""v18 ablation: decompose the persistent t256 loop's time.
Modes: 0 full | 1 -staging/-drains | 2 -ds_reads | 3 -both | 4 -barriers
| 5 MFMA+barriers only.  Only mode 0 is numerically meaningful."""
import ctypes
import pathlib
import subprocess
import sys

import torch

sys.path.insert(0, ".")
from tools.gemm_v13_screen import build  # noqa: E402

SHAPES = [("fc1_64", 67200, 3072, 768), ("square4k", 4096, 4096, 4096)]
LABELS = ["full", "-stage", "-dsread", "-both", "-barrier", "mfma+bar",
          "mb-noprio", "mfma-only", "mb-bar/2", "mb-bar/4"]


def main():
    lib = build()
    lib.cc_gemm_v18.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
        ctypes.c_long, ctypes.c_long, ctypes.c_long,
        ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_uint64,
    ]
    lib.cc_gemm_v18.restype = ctypes.c_int
    stream = torch.cuda.current_stream().cuda_stream
    ev0, ev1 = torch.cuda.Event(True), torch.cuda.Event(True)
    for label, M, N, K in SHAPES:
        torch.manual_seed(1)
        a = torch.randn(M, K).to(torch.bfloat16).cuda()
        b = torch.randn(N, K).to(torch.bfloat16).cuda()
        c = torch.empty((M, N), dtype=torch.bfloat16, device="cuda")
        flops = 2.0 * M * N * K
        out = []
        for mode in range(10):
            print(f"  mode {mode} ({LABELS[mode]})...", flush=True)
            fn = lambda: lib.cc_gemm_v18(a.data_ptr(), b.data_ptr(),
                                         c.data_ptr(), M, N, K, 1, mode, 1,
                                         stream)
            for _ in range(5):
                assert fn() == 0
            torch.cuda.synchronize()
            ev0.record()
            for _ in range(20):
                fn()
            ev1.record()
            torch.cuda.synchronize()
            us = ev0.elapsed_time(ev1) / 20 * 1e3
            out.append(f"{LABELS[mode]}={us:.0f}us({flops*1e-12/(us*1e-6):.0f}TF)")
        print(f"{label:9s} " + "  ".join(out))


if __name__ == "__main__":
    main()
