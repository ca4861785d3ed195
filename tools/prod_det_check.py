"""Determinism check of the PRODUCTION gemm at the big ViT grids (run
after any sync-structure change — the screen that caught v10)."""
import sys

import torch

sys.path.insert(0, ".")
from cosmos_curate_amd import hotpath  # noqa: E402


def main():
    lib = hotpath.require_gpu()
    stream = torch.cuda.current_stream().cuda_stream
    shapes = [(65856, 768, 3072), (67200, 2304, 768), (67200, 3072, 768),
              (67200, 768, 768), (4096, 4096, 4096), (235200, 3072, 768),
              (230496, 768, 3072)]
    bad = 0
    for (M, N, K) in shapes:
        torch.manual_seed(7)
        a = (torch.randn(M, K) * 0.3).to(torch.bfloat16).cuda()
        b = (torch.randn(N, K) * 0.3).to(torch.bfloat16).cuda()
        ref = None
        for run in range(10):
            c = torch.full((M, N), 9.0, dtype=torch.bfloat16, device="cuda")
            hotpath.check(lib.cc_gemm_bf16(a.data_ptr(), b.data_ptr(),
                                           c.data_ptr(), M, N, K, None, 1,
                                           stream))
            torch.cuda.synchronize()
            if ref is None:
                ref = c
            else:
                d = (ref != c).sum().item()
                if d:
                    bad += 1
                    print(f"NONDET {M}x{N}x{K} run{run}: {d} elems")
                    break
        # correctness vs torch on a slice
        want = (a[:128].float() @ b.float().T)
        err = (ref[:128].float() - want).abs().max().item()
        rel = err / (want.abs().max().item() + 1e-6)
        if rel > 2e-2:
            bad += 1
            print(f"WRONG {M}x{N}x{K}: rel {rel}")
    print("prod determinism+correctness:", "FAIL" if bad else "PASS")
    sys.exit(1 if bad else 0)


if __name__ == "__main__":
    main()
