// Fused bf16 LayerNorm — row-wise (x - mean)/sqrt(var + eps) * w + b.
//
// Replaces torch-rocm's vectorized_layer_norm in the ViT forward (26
// calls/step at ~2 TB/s effective, ~10% of the bench step).  Stats in
// f32 (torch layer_norm opmath semantics), bf16 in/out, f32 affine
// params.  One wave per row: H/64 elements per lane, two shfl_xor
// reduction trees (sum, sumsq), bf16x4 (8 B) loads/stores (guide §5
// rule 2: scalar/narrow bf16 access leaves >2x bandwidth on the table;
// measured 4.7 TB/s at bf16x2).
// H must be a multiple of 256 (768/1024/3072/4096 all comply).
//
// Also here: cc_embed_assemble_ln — the ViT embedding assembly
// ([CLS; patch tokens] + position embedding, then pre-LN) fused into
// one pass.  Replaces a torch cat + f32 add + bf16 cast + layer_norm
// chain (4 kernel launches, ~300 us/step at the bench shape) with one
// bandwidth-bound kernel (reference models/clip.py uses HF
// CLIPVisionEmbeddings: cat + pos_embed add, then pre_layrnorm).

#include <hip/hip_runtime.h>

#include <vector>

#include "cc_common.hpp"
#include "cc_timing.hpp"

namespace {

typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off);
  return v;
}

template <int CHUNK>  // elements per lane (H / 64), multiple of 4
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
  for (int c = 0; c < CHUNK; c += 4) {
    // coalesced: lane i reads elements (c*64 + 4i .. +3)
    bf16x4 p = *(const bf16x4*)(xr + c * 64 + 4 * lane);
#pragma unroll
    for (int j = 0; j < 4; j++) {
      vals[c + j] = (float)p[j];
      s += vals[c + j];
      ss += vals[c + j] * vals[c + j];
    }
  }
  s = wave_sum(s);
  ss = wave_sum(ss);
  const float inv_n = 1.0f / (float)H;
  const float mean = s * inv_n;
  const float var = ss * inv_n - mean * mean;
  const float rstd = rsqrtf(var + eps);
#pragma unroll
  for (int c = 0; c < CHUNK; c += 4) {
    const int i0 = c * 64 + 4 * lane;
    f32x4 wv = *(const f32x4*)(w + i0);
    f32x4 bv = *(const f32x4*)(b + i0);
    bf16x4 o;
#pragma unroll
    for (int j = 0; j < 4; j++)
      o[j] = (__bf16)((vals[c + j] - mean) * rstd * wv[j] + bv[j]);
    *(bf16x4*)(yr + i0) = o;
  }
}

// out[row] = LN( src(row) + pos[t] ) where row = f*tokens + t,
// src = cls (t==0) or tok[f*(tokens-1) + t - 1]; add in f32 like the
// torch chain it replaces (h.float() + pos -> bf16 -> layer_norm).
template <int CHUNK>
__global__ void k_embed_assemble_ln(const __bf16* __restrict__ tok,
                                    const float* __restrict__ cls,
                                    const float* __restrict__ pos,
                                    const float* __restrict__ w,
                                    const float* __restrict__ b,
                                    __bf16* __restrict__ y, long nrows,
                                    int tokens, int H, float eps) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const long row = (long)blockIdx.x * 4 + wave;
  if (row >= nrows) return;
  const long f = row / tokens;
  const int t = (int)(row - f * tokens);
  const __bf16* tr = (t == 0) ? nullptr : tok + (f * (tokens - 1) + t - 1) * H;
  const float* pr = pos + (long)t * H;
  __bf16* yr = y + row * H;

  float vals[CHUNK];
  float s = 0.0f, ss = 0.0f;
#pragma unroll
  for (int c = 0; c < CHUNK; c += 4) {
    const int i0 = c * 64 + 4 * lane;
    f32x4 pv = *(const f32x4*)(pr + i0);
#pragma unroll
    for (int j = 0; j < 4; j++) {
      // cls follows the torch path: f32 param cast to bf16 first
      float base = tr ? (float)tr[i0 + j] : (float)(__bf16)cls[i0 + j];
      // match torch: (f32 add) rounded to bf16, stats on the bf16 value
      float v = (float)(__bf16)(base + pv[j]);
      vals[c + j] = v;
      s += v;
      ss += v * v;
    }
  }
  s = wave_sum(s);
  ss = wave_sum(ss);
  const float inv_n = 1.0f / (float)H;
  const float mean = s * inv_n;
  const float var = ss * inv_n - mean * mean;
  const float rstd = rsqrtf(var + eps);
#pragma unroll
  for (int c = 0; c < CHUNK; c += 4) {
    const int i0 = c * 64 + 4 * lane;
    f32x4 wv = *(const f32x4*)(w + i0);
    f32x4 bv = *(const f32x4*)(b + i0);
    bf16x4 o;
#pragma unroll
    for (int j = 0; j < 4; j++)
      o[j] = (__bf16)((vals[c + j] - mean) * rstd * wv[j] + bv[j]);
    *(bf16x4*)(yr + i0) = o;
  }
}

// narrow rows (H=128: 2 elements/lane) keep the bf16x2 form
__global__ void k_layernorm_bf16_h128(const __bf16* __restrict__ x,
                                      const float* __restrict__ w,
                                      const float* __restrict__ b,
                                      __bf16* __restrict__ y, long M,
                                      float eps) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const long row = (long)blockIdx.x * 4 + wave;
  if (row >= M) return;
  const __bf16* xr = x + row * 128;
  __bf16* yr = y + row * 128;
  bf16x2 p = *(const bf16x2*)(xr + 2 * lane);
  float v0 = (float)p.x, v1 = (float)p.y;
  float s = wave_sum(v0 + v1);
  float ss = wave_sum(v0 * v0 + v1 * v1);
  const float mean = s / 128.0f;
  const float var = ss / 128.0f - mean * mean;
  const float rstd = rsqrtf(var + eps);
  const int i0 = 2 * lane;
  bf16x2 o;
  o.x = (__bf16)((v0 - mean) * rstd * w[i0] + b[i0]);
  o.y = (__bf16)((v1 - mean) * rstd * w[i0 + 1] + b[i0 + 1]);
  *(bf16x2*)(yr + i0) = o;
}

}  // namespace

extern "C" int cc_layernorm_bf16(const void* x, const void* w, const void* b,
                                 void* y, int64_t M, int64_t H, float eps,
                                 uint64_t stream) {
  if (!x || !w || !b || !y || M <= 0 || H <= 0)
    return cc::set_error(CC_ERR_INVALID, "bad layernorm args");
  if ((H % 256 != 0 && H != 128) || H > 64 * 64)
    return cc::set_error(CC_ERR_UNSUPPORTED,
                         "H must be 128 or k*256, <=4096 (got %lld)",
                         (long long)H);
  dim3 block(256), grid((M + 3) / 4);
  hipEvent_t ev0, ev1;
  bool timed = cc::timed_begin(stream, &ev0, &ev1);
  switch (H / 64) {
    case 2:
      hipLaunchKernelGGL(k_layernorm_bf16_h128, grid, block, 0,
                         (hipStream_t)stream, (const __bf16*)x,
                         (const float*)w, (const float*)b, (__bf16*)y,
                         (long)M, eps);
      break;
#define CASE(C)                                                            \
  case C:                                                                  \
    hipLaunchKernelGGL(k_layernorm_bf16<C>, grid, block, 0,                \
                       (hipStream_t)stream, (const __bf16*)x,              \
                       (const float*)w, (const float*)b, (__bf16*)y,       \
                       (long)M, (int)H, eps);                              \
    break;
    CASE(4) CASE(8) CASE(12) CASE(16) CASE(32) CASE(48) CASE(64)
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

extern "C" int cc_embed_assemble_ln(const void* tok, const void* cls,
                                    const void* pos, const void* w,
                                    const void* b, void* y, int64_t n_frames,
                                    int64_t tokens, int64_t H, float eps,
                                    uint64_t stream) {
  if (!tok || !cls || !pos || !w || !b || !y || n_frames <= 0 || tokens <= 1 ||
      H <= 0)
    return cc::set_error(CC_ERR_INVALID, "bad embed_assemble args");
  if (H % 256 != 0 || H > 64 * 64)
    return cc::set_error(CC_ERR_UNSUPPORTED, "H must be k*256, <=4096 (got %lld)",
                         (long long)H);
  const long nrows = (long)n_frames * tokens;
  dim3 block(256), grid((nrows + 3) / 4);
  hipEvent_t ev0, ev1;
  bool timed = cc::timed_begin(stream, &ev0, &ev1);
  switch (H / 64) {
#define CASE(C)                                                            \
  case C:                                                                  \
    hipLaunchKernelGGL(k_embed_assemble_ln<C>, grid, block, 0,             \
                       (hipStream_t)stream, (const __bf16*)tok,            \
                       (const float*)cls, (const float*)pos,               \
                       (const float*)w, (const float*)b, (__bf16*)y,       \
                       nrows, (int)tokens, (int)H, eps);                   \
    break;
    CASE(4) CASE(8) CASE(12) CASE(16) CASE(32) CASE(48) CASE(64)
#undef CASE
    default:
      return cc::set_error(CC_ERR_UNSUPPORTED, "unsupported H %lld", (long long)H);
  }
  hipError_t e = hipGetLastError();
  if (timed) cc::timed_end("embed_assemble_ln", stream, ev0, ev1);
  if (e != hipSuccess)
    return cc::set_error(CC_ERR_HIP, "embed_assemble launch: %s",
                         hipGetErrorString(e));
  return CC_OK;
}
