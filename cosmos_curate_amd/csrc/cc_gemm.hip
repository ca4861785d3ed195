// bf16 MFMA GEMM for the ViT forward — hand-written for gfx950 (CDNA4).
//
// Replaces the cuBLAS GEMMs under CLIPModel.get_image_features
// (/root/reference/cosmos_curate/models/clip.py:71): patch-embed-as-GEMM,
// QKV / attention-out projections, MLP fc1/fc2, visual projection
// (SURVEY.md §2b row 8).
//
// C[M,N] = A[M,K] x B[N,K]^T (+ bias[N]); A,B bf16 row-major, accumulate
// f32 on v_mfma_f32_16x16x32_bf16, output f32 or bf16.
//
// Design (cdna_hip_programming.md §5 "canonical CDNA GEMM", step-3 ladder
// structure):
//   - 128x128 block tile, BK=64, 256 threads = 4 waves in a 2x2 wave grid,
//     each wave owns a 64x64 sub-tile = 4x4 MFMA fragments of 16x16.
//   - global->LDS staging via __builtin_amdgcn_global_load_lds width 16
//     (the compiler never auto-emits it), double-buffered, one barrier +
//     one vmcnt(0) per K-tile.
//   - "weight layout" B[N,K] makes both operands K-major: one staging
//     scheme, coalesced 128-byte rows, K must be a multiple of 64 (all ViT
//     shapes are: 768/2304/3072/512).
//   - M and N edges handled by clamping the *global* load address (garbage
//     lands only in masked-out output rows/cols) and predicated stores.
//
// Numerics: f32 accumulation in k-order within each K-tile (MFMA chains);
// parity vs torch fp32 matmul is checked at cosine/rtol level
// (tests/test_gpu_vit.py), not bitwise — bf16 inputs round first.

#include <hip/hip_runtime.h>

#include "cc_common.hpp"

#define CC_CHECK_HIP(expr)                                                   \
  do {                                                                       \
    hipError_t _e = (expr);                                                  \
    if (_e != hipSuccess)                                                    \
      return cc::set_error(CC_ERR_HIP, "%s failed: %s", #expr,               \
                           hipGetErrorString(_e));                           \
  } while (0)

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int WAVES_M = 2, WAVES_N = 2;  // wave grid
constexpr int WM = BM / WAVES_M;         // 64 rows per wave
constexpr int WN = BN / WAVES_N;         // 64 cols per wave
constexpr int FRAG = 16;                 // mfma 16x16x32
constexpr int MFR = WM / FRAG;           // 4 m-fragments
constexpr int NFR = WN / FRAG;           // 4 n-fragments

__device__ __forceinline__ unsigned short f32_to_bf16_rne(float v) {
  union {
    float f;
    unsigned int u;
  } cv{v};
  unsigned int lsb = (cv.u >> 16) & 1;
  return (unsigned short)((cv.u + 0x7fffu + lsb) >> 16);
}

// stage one 32-row slice of a [rows x BK] bf16 tile into LDS via glds.
// lds_base: wave-uniform LDS address of this wave's 32-row slice.
// src: global base of the matrix (bf16), ld = row stride in elements.
// row0: first global row of the slice; nrows_clamp: clamp rows to [0, limit).
__device__ __forceinline__ void stage_slice(const __bf16* __restrict__ src,
                                            long ld, long row0, long row_limit,
                                            long k0, __bf16* lds_base,
                                            int lane) {
#pragma unroll
  for (int j = 0; j < 4; j++) {  // 4 x 1KB chunks = 32 rows
    long grow = row0 + j * 8 + (lane >> 3);
    grow = grow < 0 ? 0 : (grow >= row_limit ? row_limit - 1 : grow);
    const __bf16* gptr = src + grow * ld + k0 + (long)(lane & 7) * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)gptr,
        (__attribute__((address_space(3))) unsigned int*)(lds_base + j * 8 * BK),
        16, 0, 0);
  }
}

__global__ __launch_bounds__(256, 2) void k_gemm_bf16(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ C, const float* __restrict__ bias, long M, long N,
    long K, int c_is_bf16) {
  __shared__ __bf16 lds[2 * (BM + BN) * BK];  // As[2][128][64], Bs[2][128][64]
  // single __shared__ object (G16 trap 4a); buffer b lives at:
  //   A: lds + b*BM*BK            B: lds + 2*BM*BK + b*BN*BK
#define AS(b) (lds + (b) * (BM * BK))
#define BS(b) (lds + 2 * (BM * BK) + (b) * (BN * BK))

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int waveM = wid >> 1, waveN = wid & 1;

  const long bm = (long)blockIdx.y * BM;
  const long bn = (long)blockIdx.x * BN;

  // this wave stages rows [32*wid, 32*wid+32) of both tiles
  const long arow0 = bm + 32 * wid;
  const long brow0 = bn + 32 * wid;

  f32x4 acc[MFR][NFR] = {};

  const long KT = K / BK;
  // prologue: stage tile 0 into buffer 0
  stage_slice(A, K, arow0, M, 0, AS(0) + 32 * wid * BK, lane);
  stage_slice(B, K, brow0, N, 0, BS(0) + 32 * wid * BK, lane);

  int buf = 0;
  for (long kt = 0; kt < KT; ++kt) {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    if (kt + 1 < KT) {
      const long k0 = (kt + 1) * BK;
      stage_slice(A, K, arow0, M, k0, AS(buf ^ 1) + 32 * wid * BK, lane);
      stage_slice(B, K, brow0, N, k0, BS(buf ^ 1) + 32 * wid * BK, lane);
    }
    // compute on buf: 2 k-steps of 32, 16 MFMA each
    const __bf16* At = AS(buf);
    const __bf16* Bt = BS(buf);
    const int arow_frag = waveM * WM + (lane & 15);
    const int brow_frag = waveN * WN + (lane & 15);
    const int koff = 8 * (lane >> 4);
#pragma unroll
    for (int kk = 0; kk < BK; kk += 32) {
      bf16x8 afrag[MFR], bfrag[NFR];
#pragma unroll
      for (int m = 0; m < MFR; m++)
        afrag[m] = *(const bf16x8*)(At + (arow_frag + m * FRAG) * BK + kk + koff);
#pragma unroll
      for (int n = 0; n < NFR; n++)
        bfrag[n] = *(const bf16x8*)(Bt + (brow_frag + n * FRAG) * BK + kk + koff);
#pragma unroll
      for (int m = 0; m < MFR; m++)
#pragma unroll
        for (int n = 0; n < NFR; n++)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[m], bfrag[n], acc[m][n], 0, 0, 0);
    }
    buf ^= 1;
  }

  // epilogue: C/D map for 16x16x32: col = lane&15, row = (lane>>4)*4 + reg
  const long crow_base = bm + waveM * WM + 4 * (lane >> 4);
  const long ccol_base = bn + waveN * WN + (lane & 15);
#pragma unroll
  for (int m = 0; m < MFR; m++) {
#pragma unroll
    for (int n = 0; n < NFR; n++) {
      const long col = ccol_base + n * FRAG;
      if (col >= N) continue;
      const float bval = bias ? bias[col] : 0.0f;
#pragma unroll
      for (int r = 0; r < 4; r++) {
        const long row = crow_base + m * FRAG + r;
        if (row >= M) continue;
        float v = acc[m][n][r] + bval;
        if (c_is_bf16)
          ((unsigned short*)C)[row * N + col] = f32_to_bf16_rne(v);
        else
          ((float*)C)[row * N + col] = v;
      }
    }
  }
}

inline void record_timing(const char* name, float ms) {
  auto& ts = cc::timing();
  std::lock_guard<std::mutex> lk(ts.mu);
  auto& e = ts.entries[name];
  e.total_ms += ms;
  e.count += 1;
}

}  // namespace

extern "C" int cc_gemm_bf16(const void* A, const void* B, void* C, int64_t M,
                            int64_t N, int64_t K, const float* bias,
                            int c_dtype, uint64_t stream) {
  if (!A || !B || !C || M <= 0 || N <= 0 || K <= 0)
    return cc::set_error(CC_ERR_INVALID, "bad gemm args");
  if (K % BK != 0)
    return cc::set_error(CC_ERR_UNSUPPORTED,
                         "cc_gemm_bf16 requires K %% 64 == 0 (got %lld)",
                         (long long)K);
  dim3 block(256);
  dim3 grid((N + BN - 1) / BN, (M + BM - 1) / BM);
  auto& ts = cc::timing();
  hipEvent_t ev0 = nullptr, ev1 = nullptr;
  bool timed = false;
  if (ts.enabled) {
    if (hipEventCreate(&ev0) == hipSuccess && hipEventCreate(&ev1) == hipSuccess) {
      hipEventRecord(ev0, (hipStream_t)stream);
      timed = true;
    }
  }
  hipLaunchKernelGGL(k_gemm_bf16, grid, block, 0, (hipStream_t)stream,
                     (const __bf16*)A, (const __bf16*)B, C, bias, (long)M,
                     (long)N, (long)K, c_dtype == 1 ? 1 : 0);
  hipError_t e = hipGetLastError();
  if (timed) {
    hipEventRecord(ev1, (hipStream_t)stream);
    hipEventSynchronize(ev1);
    float ms = 0;
    hipEventElapsedTime(&ms, ev0, ev1);
    record_timing("gemm_bf16", ms);
    hipEventDestroy(ev0);
    hipEventDestroy(ev1);
  }
  if (e != hipSuccess)
    return cc::set_error(CC_ERR_HIP, "gemm launch: %s", hipGetErrorString(e));
  return CC_OK;
}
