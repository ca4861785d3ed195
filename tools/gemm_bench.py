"""Standalone microbench for cc_gemm_bf16 on the ViT shapes.

Per shape: warmup + K timed iterations with HIP events (library timing),
reports TF/s and the fraction of the 2.5 PF dense bf16 peak.  Used to
iterate on the GEMM kernel without paying for a full bench run.

    gpurun -- 'python tools/gemm_bench.py'
"""

from __future__ import annotations

import sys

import torch

sys.path.insert(0, ".")

from cosmos_curate_amd import hotpath  # noqa: E402

SHAPES = [
    # (label, M, N, K) — F=336 frames (16 clips x 21)
    ("patch", 336 * 49, 768, 3072),
    ("qkv", 336 * 50, 2304, 768),
    ("attn_out", 336 * 50, 768, 768),
    ("fc1", 336 * 50, 3072, 768),
    ("fc2", 336 * 50, 768, 3072),
    ("proj", 336, 512, 768),
    ("square4k", 4096, 4096, 4096),
    ("square8k", 8192, 8192, 8192),
]


def main() -> None:
    lib = hotpath.require_gpu()
    stream = torch.cuda.current_stream().cuda_stream
    print(f"{'shape':10s} {'M':>6s} {'N':>5s} {'K':>5s} {'TF/s':>8s} {'%peak':>6s}")
    for label, M, N, K in SHAPES:
        torch.manual_seed(1)
        a = torch.randn(M, K).to(torch.bfloat16).cuda()
        b = torch.randn(N, K).to(torch.bfloat16).cuda()
        c = torch.empty((M, N), dtype=torch.bfloat16, device="cuda")
        for _ in range(5):
            hotpath.check(lib.cc_gemm_bf16(a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, None, 1, stream))
        torch.cuda.synchronize()
        hotpath.timing_enable(True)
        iters = 30 if M * N * K < 2**36 else 10
        for _ in range(iters):
            hotpath.check(lib.cc_gemm_bf16(a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K, None, 1, stream))
        torch.cuda.synchronize()
        ms, cnt = hotpath.timing_report("gemm_bf16")
        hotpath.timing_enable(False)
        tf = 2.0 * M * N * K * cnt / (ms / 1e3) / 1e12
        print(f"{label:10s} {M:6d} {N:5d} {K:5d} {tf:8.1f} {tf/2500*100:5.1f}%")
        # correctness spot check vs torch (small slice)
        want = (a[:64].float() @ b[:64].float().T).cpu()
        got = c[:64, :64].float().cpu()
        err = (got - want[:, :64]).abs().max().item()
        scale = want.abs().max().item() + 1e-6
        assert err / scale < 2e-2, f"{label}: rel err {err/scale}"


if __name__ == "__main__":
    main()
