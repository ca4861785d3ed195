"""Race screen for the v10 8-phase sync structure (guide 'two-lane
discipline': sync-structure edits are NEW templates -> multi-run
refcheck at several sizes vs the production kernel)."""
import ctypes
import pathlib
import sys

import torch

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
from cosmos_curate_amd import hotpath  # noqa: E402
from tools.gemm_bench2 import build  # noqa: E402


def main():
    prod = hotpath.require_gpu()
    _, _, var10 = build()
    stream = torch.cuda.current_stream().cuda_stream
    shapes = [(256, 256, 256), (512, 512, 512), (512, 256, 768),
              (4096, 4096, 4096), (1000, 777, 768), (260, 300, 384)]
    bad = 0
    for rep in range(5):
        for (M, N, K) in shapes:
            torch.manual_seed(rep * 100 + M)
            a = (torch.randn(M, K) * 0.3).to(torch.bfloat16).cuda()
            b = (torch.randn(N, K) * 0.3).to(torch.bfloat16).cuda()
            c1 = torch.empty((M, N), dtype=torch.bfloat16, device="cuda")
            c2 = torch.full((M, N), 7.0, dtype=torch.bfloat16, device="cuda")
            hotpath.check(prod.cc_gemm_bf16(a.data_ptr(), b.data_ptr(),
                                            c1.data_ptr(), M, N, K, None, 1, stream))
            rc = var10.cc_gemm_v10(a.data_ptr(), b.data_ptr(), c2.data_ptr(),
                                   M, N, K, 1, stream)
            assert rc == 0, hotpath.last_error()
            torch.cuda.synchronize()
            err = (c1.float() - c2.float()).abs().max().item()
            if err > 1e-3:
                bad += 1
                print(f"MISMATCH rep{rep} {M}x{N}x{K}: max err {err}")
    print("race screen:", "FAIL" if bad else "PASS", f"({5 * len(shapes)} runs)")


if __name__ == "__main__":
    main()
