"""Tight loop of the production GEMM with BENCH-faithful epilogues for
PMC passes (fc1: act+bias; attn_out: bias+residual)."""
import os
import pathlib
import sys

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
import torch  # noqa: E402

from cosmos_curate_amd import hotpath  # noqa: E402

lib = hotpath.require_gpu()
stream = torch.cuda.current_stream().cuda_stream
shapes = {
    "fc1": [(67200, 3072, 768, 1, 0)],     # act=1 (quick-gelu), no res
    "out": [(67200, 768, 768, 0, 1)],      # residual fused
    "bare_fc1": [(67200, 3072, 768, 0, -1)],  # no bias, no res (screen cfg)
    "fc1_b160": [(168000, 3072, 768, 1, 0)],  # bench default batch 160
    "out_b160": [(168000, 768, 768, 0, 1)],
}
for (M, N, K, act, res) in shapes[os.environ.get("CC_PMC_SHAPE", "fc1")]:
    torch.manual_seed(1)
    a = (torch.randn(M, K) * 0.3).to(torch.bfloat16).cuda()
    b = (torch.randn(N, K) * 0.3).to(torch.bfloat16).cuda()
    bias = torch.randn(N).float().cuda()
    r = (torch.randn(M, N) * 0.3).to(torch.bfloat16).cuda() if res == 1 else None
    c = torch.empty((M, N), dtype=torch.bfloat16, device="cuda")
    for _ in range(20):
        hotpath.check(lib.cc_gemm_bf16_ex(
            a.data_ptr(), b.data_ptr(), c.data_ptr(), M, N, K,
            None if res == -1 else bias.data_ptr(), 1, act,
            r.data_ptr() if r is not None else None, stream))
    torch.cuda.synchronize()
print("done")
