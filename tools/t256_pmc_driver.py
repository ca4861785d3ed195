"""Tight loop of the production GEMM at two t256 shapes for PMC passes."""
import pathlib
import sys

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
import torch  # noqa: E402

from cosmos_curate_amd import hotpath  # noqa: E402

lib = hotpath.require_gpu()
stream = torch.cuda.current_stream().cuda_stream
import os
shapes = {'qkv': [(67200, 2304, 768)], 'fc1': [(67200, 3072, 768)], 'sq': [(8192, 8192, 8192)], 'both': [(67200, 2304, 768), (8192, 8192, 8192)]}
for (M, N, K) in shapes[os.environ.get('CC_PMC_SHAPE', 'both')]:
    torch.manual_seed(1)
    a = (torch.randn(M, K) * 0.3).to(torch.bfloat16).cuda()
    b = (torch.randn(N, K) * 0.3).to(torch.bfloat16).cuda()
    c = torch.empty((M, N), dtype=torch.bfloat16, device="cuda")
    for _ in range(20):
        hotpath.check(lib.cc_gemm_bf16(a.data_ptr(), b.data_ptr(), c.data_ptr(),
                                       M, N, K, None, 1, stream))
    torch.cuda.synchronize()
print("done")
