// Fused bf16 LayerNorm — row-wise (x - mean)/sqrt(var + eps) * w + b.
//
// Replaces torch-rocm's vectorized_layer_norm in the ViT forward (26
// calls/step at ~2 TB/s effective, ~10% of the bench step).  Stats in
// f32 (torch layer_norm opmath semantics), bf16 in/out, f32 affine
// params.  One wave per row: H/64 elements per lane, two shfl_xor
// reduction trees (sum, sumsq), coalesced bf16x2 loads/stores.
// H must be a multiple of 128 (768/1024/3072/4096 all comply).

#include <hip/hip_runtime.h>

#include <vector>

#include "cc_common.hpp"
#include "cc_timing.hpp"

namespace {

typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2;

__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off);
  return v;
}

template <int CHUNK>  // elements per lane (H / 64)
__global__ void k_layernorm_bf16(const __bf16* __restrict__ x,
                                 const float* __restrict__ w,
                                 const float* __restrict__ b,
                                 __bf16* __restrict__ y, long M, int H,
                                 float eps) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;          // 4 waves per block
  const long row = (long)blockIdx.x * 4 + wave;
  if (row >= M) return;
  const __bf16* xr = x + row * H;
  __bf16* yr = y + row * H;

  float vals[CHUNK];
  float s = 0.0f, ss = 0.0f;
#pragma unroll
  for (int c = 0; c < CHUNK; c += 2) {
    // coalesced: lane i reads elements (c*64 + 2i, +1)
    bf16x2 p = *(const bf16x2*)(xr + c * 64 + 2 * lane);
    vals[c] = (float)p.x;
    vals[c + 1] = (float)p.y;
    s += vals[c] + vals[c + 1];
    ss += vals[c] * vals[c] + vals[c + 1] * vals[c + 1];
  }
  s = wave_sum(s);
  ss = wave_sum(ss);
  const float inv_n = 1.0f / (float)H;
  const float mean = s * inv_n;
  const float var = ss * inv_n - mean * mean;
  const float rstd = rsqrtf(var + eps);
#pragma unroll
  for (int c = 0; c < CHUNK; c += 2) {
    const int i0 = c * 64 + 2 * lane;
    float w0 = w[i0], w1 = w[i0 + 1];
    float b0 = b[i0], b1 = b[i0 + 1];
    bf16x2 o;
    o.x = (__bf16)((vals[c] - mean) * rstd * w0 + b0);
    o.y = (__bf16)((vals[c + 1] - mean) * rstd * w1 + b1);
    *(bf16x2*)(yr + i0) = o;
  }
}

}  // namespace

extern "C" int cc_layernorm_bf16(const void* x, const void* w, const void* b,
                                 void* y, int64_t M, int64_t H, float eps,
                                 uint64_t stream) {
  if (!x || !w || !b || !y || M <= 0 || H <= 0)
    return cc::set_error(CC_ERR_INVALID, "bad layernorm args");
  if (H % 128 != 0 || H > 64 * 64)
    return cc::set_error(CC_ERR_UNSUPPORTED, "H must be k*128, <=4096 (got %lld)",
                         (long long)H);
  dim3 block(256), grid((M + 3) / 4);
  hipEvent_t ev0, ev1;
  bool timed = cc::timed_begin(stream, &ev0, &ev1);
  switch (H / 64) {
#define CASE(C)                                                            \
  case C:                                                                  \
    hipLaunchKernelGGL(k_layernorm_bf16<C>, grid, block, 0,                \
                       (hipStream_t)stream, (const __bf16*)x,              \
                       (const float*)w, (const float*)b, (__bf16*)y,       \
                       (long)M, (int)H, eps);                              \
    break;
    CASE(2) CASE(4) CASE(6) CASE(8) CASE(12) CASE(16) CASE(32) CASE(48) CASE(64)
#undef CASE
    default:
      return cc::set_error(CC_ERR_UNSUPPORTED, "unsupported H %lld", (long long)H);
  }
  hipError_t e = hipGetLastError();
  if (timed) cc::timed_end("layernorm_bf16", stream, ev0, ev1);
  if (e != hipSuccess)
    return cc::set_error(CC_ERR_HIP, "layernorm launch: %s", hipGetErrorString(e));
  return CC_OK;
}
