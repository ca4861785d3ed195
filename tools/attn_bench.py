"""attn_small microbench at the bench shape (b224: 4704 frames, seq 50)."""
import ctypes
import sys

import torch

sys.path.insert(0, ".")
from cosmos_curate_amd import hotpath  # noqa: E402


def main():
    lib = hotpath.require_gpu()
    stream = torch.cuda.current_stream().cuda_stream
    n, seq, heads, hidden = 4704, 50, 12, 768
    torch.manual_seed(1)
    qkv = (torch.randn(n * seq, 3 * hidden) * 0.3).to(torch.bfloat16).cuda()
    out = torch.empty((n * seq, hidden), dtype=torch.bfloat16, device="cuda")
    scale = 0.125
    call = lambda: hotpath.check(lib.cc_attn_small(
        qkv.data_ptr(), out.data_ptr(), n, seq, heads, hidden,
        ctypes.c_float(scale), stream))
    for _ in range(5):
        call()
    torch.cuda.synchronize()
    ev0, ev1 = torch.cuda.Event(True), torch.cuda.Event(True)
    ev0.record()
    for _ in range(30):
        call()
    ev1.record()
    torch.cuda.synchronize()
    us = ev0.elapsed_time(ev1) / 30 * 1e3
    traffic = n * heads * (3 * seq * 64 + seq * 64) * 2  # bytes (in+out)
    print(f"attn_small b224: {us:.0f} us/launch, {traffic/us/1e6:.2f} TB/s")
    # correctness vs sdpa on a slice
    import math
    q, k, v = (qkv.reshape(n, seq, 3, heads, 64)[:2, :, i].permute(0, 2, 1, 3).float()
               for i in range(3))
    want = torch.nn.functional.scaled_dot_product_attention(q, k, v, scale=scale)
    want = want.permute(0, 2, 1, 3).reshape(2 * seq, hidden)
    torch.testing.assert_close(out[:2 * seq].float(), want, rtol=2e-2, atol=2e-2)
    print("parity OK")


if __name__ == "__main__":
    main()
