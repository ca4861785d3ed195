// V10: the 256x256 8-phase GEMM template (cdna_hip_programming.md
// "The 256² 8-phase template"), rebuilt with this repo's LDS image
// (row-major [rows][BK] bf16, slot = k16 ^ (row&7) swizzle).
//
// Geometry: 512 threads = 8 waves as 2(M) x 4(N); per-wave output
// 128x64 = 8x4 fragments of 16x16; BK=64; LDS = 2 buf x (A 32KB + B
// 32KB) = 128 KiB; 1 WG/CU.
//
// Phase machine per iteration (2 K-tiles t, t+1), derived so every
// counted wait is safe by construction (issue order in comments):
//   ph1: ds_read B(t) all 4 N-frags x 2k (8) + A quadrant0 (4);
//        glds A(t+1) half0                      [A-buf((t+1)&1) free]
//   ph2: ds_read A q1; glds A(t+1) half1
//   ph3: ds_read A q2; s_waitcnt vmcnt(0) (full drain: everything ph5
//        needs is already issued; counted waits were measurably unsound
//        here — VMEM completions retire out of order under load, see
//        profiles/r01_t256_det2.log); then glds B(t+2) half0
//        [B-buf(t&1) free after ph1]
//   ph4: ds_read A q3; glds B(t+2) half1  (B(t+2) spans the barrier)
//   ph5-8: same with t+1, staging A(t+2) h0,h1 then B(t+3) h0,h1.
// Each phase: [reads; glds] -> s_barrier -> lgkmcnt(0) -> setprio(1)
// -> 16 MFMA -> setprio(0) -> s_barrier (raw barriers: no vmcnt(0)
// glds drain, the template's whole point).
//
// Verification lanes (guide "two-lane discipline"): refcheck vs the
// production kernel at 256/512/4096 + multi-run race screen (sync
// structure is new) in tools/gemm_bench2.py / test_gpu_kernels.

#include <hip/hip_runtime.h>

#include "../cosmos_curate_amd/csrc/cc_common.hpp"

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

constexpr int BM = 256, BN = 256, BK = 64;
constexpr int WM = 128, WN = 64;  // per-wave output (2x4 wave grid)
constexpr int FRAG = 16;
constexpr int MFR = WM / FRAG;  // 8
constexpr int NFR = WN / FRAG;  // 4

__device__ __forceinline__ unsigned short bf16_rne(float v) {
  union { float f; unsigned int u; } cv{v};
  return (unsigned short)((cv.u + 0x7fffu + ((cv.u >> 16) & 1)) >> 16);
}

// one wave stages its 16-row slice of a 128-row half-tile: 2 glds.
// image layout [rows][BK] bf16, swizzled slot = gk16 ^ (row&7).
__device__ __forceinline__ void stage_half(const __bf16* __restrict__ src,
                                           long ld, long row0,
                                           long row_limit, long k0,
                                           __bf16* lds_rowbase, int lane) {
  const int lrow8 = lane >> 3;
  const int slot = lane & 7;
  const int gk16 = slot ^ lrow8;
#pragma unroll
  for (int j = 0; j < 2; j++) {
    long grow = row0 + j * 8 + lrow8;
    grow = grow < 0 ? 0 : (grow >= row_limit ? row_limit - 1 : grow);
    const __bf16* gptr = src + grow * ld + k0 + (long)gk16 * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)gptr,
        (__attribute__((address_space(3))) unsigned int*)(lds_rowbase +
                                                          j * 8 * BK),
        16, 0, 0);
  }
}

__device__ __forceinline__ bf16x8 frag_read(const __bf16* tile, int row,
                                            int k16) {
  int slot = k16 ^ (row & 7);
  return *(const bf16x8*)(tile + (long)row * BK + slot * 8);
}

template <bool TWO_BARRIERS>
__global__ __launch_bounds__(512, 1) void k_gemm_v10(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ C, long M, long N, long K, int c_is_bf16) {
  __shared__ __bf16 lds[2 * (BM + BN) * BK];  // 128 KiB
#define A10(b) (lds + (b) * (BM * BK))
#define B10(b) (lds + 2 * (BM * BK) + (b) * (BN * BK))
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int waveM = wid >> 2, waveN = wid & 3;
  const long bm = (long)blockIdx.y * BM;
  const long bn = (long)blockIdx.x * BN;

  f32x4 acc[MFR][NFR] = {};
  const long KT = K / BK;  // caller guarantees KT even, >= 4

  // wave's 16-row staging slice inside a 128-row half-tile
  const long srow = wid * 16;

#define STAGE_A(t, h)                                                       \
  stage_half(A, K, bm + (h) * 128 + srow, M, (t) * BK,                      \
             A10((t) & 1) + ((h) * 128 + srow) * BK, lane)
#define STAGE_B(t, h)                                                       \
  stage_half(B, K, bn + (h) * 128 + srow, N, (t) * BK,                      \
             B10((t) & 1) + ((h) * 128 + srow) * BK, lane)

  // prologue: A(t0) h0,h1, B(t0) h0,h1, B(t1) h0,h1  (12 instructions);
  // wait for t0's 8 (the 4 newest = B(t1)) then publish.
  STAGE_A(0, 0);
  STAGE_A(0, 1);
  STAGE_B(0, 0);
  STAGE_B(0, 1);
  STAGE_B(1, 0);
  STAGE_B(1, 1);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  const int arow_base = waveM * WM + (lane & 15);
  const int brow_base = waveN * WN + (lane & 15);

  bf16x8 bfragT[NFR][2];  // held for the whole K-tile
  bf16x8 afrag[2][2];     // one quadrant: 2 M-frags x 2 k-steps

  // one phase: reads were issued by the caller; this does the sync +
  // MFMA block for quadrant q against bfragT.
#define PHASE_MFMA(q)                                                       \
  do {                                                                      \
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");                      \
    __builtin_amdgcn_s_setprio(1);                                          \
    _Pragma("unroll") for (int g = 0; g < 2; g++) {                         \
      _Pragma("unroll") for (int m = 0; m < 2; m++) {                       \
        _Pragma("unroll") for (int n = 0; n < NFR; n++) {                   \
          acc[2 * (q) + m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(    \
              afrag[m][g], bfragT[n][g], acc[2 * (q) + m][n], 0, 0, 0);     \
        }                                                                   \
      }                                                                     \
    }                                                                       \
    __builtin_amdgcn_s_setprio(0);                                          \
    if constexpr (TWO_BARRIERS) __builtin_amdgcn_s_barrier();               \
  } while (0)

#define READ_A(At, q)                                                       \
  _Pragma("unroll") for (int g = 0; g < 2; g++) {                           \
    const int k16 = (g << 2) + (lane >> 4);                                 \
    _Pragma("unroll") for (int m = 0; m < 2; m++) afrag[m][g] =             \
        frag_read(At, arow_base + (q) * 32 + m * FRAG, k16);                \
  }

#define READ_B(Bt)                                                          \
  _Pragma("unroll") for (int g = 0; g < 2; g++) {                           \
    const int k16 = (g << 2) + (lane >> 4);                                 \
    _Pragma("unroll") for (int n = 0; n < NFR; n++) bfragT[n][g] =          \
        frag_read(Bt, brow_base + n * FRAG, k16);                           \
  }

  for (long it = 0; it < KT / 2; ++it) {
    const long t = 2 * it;
    {
      const __bf16* At = A10(t & 1);
      const __bf16* Bt = B10(t & 1);
      // ph1
      READ_B(Bt);
      READ_A(At, 0);
      if (t + 1 < KT) STAGE_A(t + 1, 0);
      __builtin_amdgcn_s_barrier();
      PHASE_MFMA(0);
      // ph2
      READ_A(At, 1);
      if (t + 1 < KT) STAGE_A(t + 1, 1);
      __builtin_amdgcn_s_barrier();
      PHASE_MFMA(1);
      // ph3
      READ_A(At, 2);
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // see prod note
      if (t + 2 < KT) STAGE_B(t + 2, 0);
      __builtin_amdgcn_s_barrier();
      PHASE_MFMA(2);
      // ph4
      READ_A(At, 3);
      if (t + 2 < KT) STAGE_B(t + 2, 1);
      __builtin_amdgcn_s_barrier();
      PHASE_MFMA(3);
    }
    {
      const long u = t + 1;
      const __bf16* At = A10(u & 1);
      const __bf16* Bt = B10(u & 1);
      // ph5
      READ_B(Bt);
      READ_A(At, 0);
      if (u + 1 < KT) STAGE_A(u + 1, 0);
      __builtin_amdgcn_s_barrier();
      PHASE_MFMA(0);
      // ph6
      READ_A(At, 1);
      if (u + 1 < KT) STAGE_A(u + 1, 1);
      __builtin_amdgcn_s_barrier();
      PHASE_MFMA(1);
      // ph7
      READ_A(At, 2);
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      if (u + 2 < KT) STAGE_B(u + 2, 0);
      __builtin_amdgcn_s_barrier();
      PHASE_MFMA(2);
      // ph8
      READ_A(At, 3);
      if (u + 2 < KT) STAGE_B(u + 2, 1);
      __builtin_amdgcn_s_barrier();
      PHASE_MFMA(3);
    }
  }

  // epilogue: same C/D map as the production 16x16 body
  const long crow_base = bm + waveM * WM + 4 * (lane >> 4);
  const long ccol_base = bn + waveN * WN + (lane & 15);
#pragma unroll
  for (int m = 0; m < MFR; m++) {
#pragma unroll
    for (int n = 0; n < NFR; n++) {
      const long col = ccol_base + n * FRAG;
      if (col >= N) continue;
#pragma unroll
      for (int r = 0; r < 4; r++) {
        const long row = crow_base + m * FRAG + r;
        if (row >= M) continue;
        float v = acc[m][n][r];
        if (c_is_bf16)
          ((unsigned short*)C)[row * N + col] = bf16_rne(v);
        else
          ((float*)C)[row * N + col] = v;
      }
    }
  }
#undef A10
#undef B10
#undef STAGE_A
#undef STAGE_B
#undef PHASE_MFMA
#undef READ_A
#undef READ_B
}

}  // namespace

extern "C" int cc_gemm_v10(const void* A, const void* B, void* C, int64_t M,
                           int64_t N, int64_t K, int c_dtype,
                           uint64_t stream) {
  if (K % (2 * BK) != 0 || K / BK < 4)
    return cc::set_error(CC_ERR_UNSUPPORTED, "K must be multiple of 128, >=256");
  dim3 block(512);
  dim3 grid((N + BN - 1) / BN, (M + BM - 1) / BM);
  hipLaunchKernelGGL((k_gemm_v10<true>), grid, block, 0, (hipStream_t)stream,
                     (const __bf16*)A, (const __bf16*)B, C, (long)M, (long)N,
                     (long)K, c_dtype == 1);
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) return cc::set_error(CC_ERR_HIP, "%s", hipGetErrorString(e));
  return CC_OK;
}


// ---- V12: v11 + A-quadrant reads software-pipelined into the MFMA
// region.  afrag is double-buffered (+16 VGPR): the reads for quadrant
// q+1 are issued textually after MFMA(q) whose operands live in the
// OTHER afrag set, so the scheduler interleaves them under the MFMAs
// and the next phase's lgkmcnt(0) is nearly free.  B reads and glds
// keep the v11 placement.  Sync structure: same barriers/drains as v11.
namespace v12 {
constexpr int WM = 128, WN = 64, FRAG = 16;
constexpr int MFR = WM / FRAG, NFR = WN / FRAG;

__global__ __launch_bounds__(512, 1) void k_gemm_v12(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ C, long M, long N, long K, int c_is_bf16) {
  __shared__ __bf16 lds[2 * (BM + BN) * BK];
#define A12(b) (lds + (b) * (BM * BK))
#define B12(b) (lds + 2 * (BM * BK) + (b) * (BN * BK))
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int waveM = wid >> 2, waveN = wid & 3;
  const long bm = (long)blockIdx.y * BM;
  const long bn = (long)blockIdx.x * BN;

  f32x4 acc[MFR][NFR] = {};
  const long KT = K / BK;
  const long srow = wid * 16;

#define STG_A(t, h)                                                         \
  stage_half(A, K, bm + (h) * 128 + srow, M, (t) * BK,                      \
             A12((t) & 1) + ((h) * 128 + srow) * BK, lane)
#define STG_B(t, h)                                                         \
  stage_half(B, K, bn + (h) * 128 + srow, N, (t) * BK,                      \
             B12((t) & 1) + ((h) * 128 + srow) * BK, lane)

  STG_A(0, 0);
  STG_A(0, 1);
  STG_B(0, 0);
  STG_B(0, 1);
  STG_B(1, 0);
  STG_B(1, 1);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  const int arow_base = waveM * WM + (lane & 15);
  const int brow_base = waveN * WN + (lane & 15);
  bf16x8 bfragT[NFR][2];
  bf16x8 afrag[2][2][2];  // [set][m][g]

#define RD_A(At, q, set)                                                    \
  _Pragma("unroll") for (int g = 0; g < 2; g++) {                           \
    const int k16 = (g << 2) + (lane >> 4);                                 \
    _Pragma("unroll") for (int m = 0; m < 2; m++) afrag[set][m][g] =        \
        frag_read(At, arow_base + (q) * 32 + m * FRAG, k16);                \
  }
#define RD_B(Bt)                                                            \
  _Pragma("unroll") for (int g = 0; g < 2; g++) {                           \
    const int k16 = (g << 2) + (lane >> 4);                                 \
    _Pragma("unroll") for (int n = 0; n < NFR; n++) bfragT[n][g] =          \
        frag_read(Bt, brow_base + n * FRAG, k16);                           \
  }
#define MM(q, set)                                                          \
  _Pragma("unroll") for (int g = 0; g < 2; g++)                             \
  _Pragma("unroll") for (int m = 0; m < 2; m++)                             \
  _Pragma("unroll") for (int n = 0; n < NFR; n++)                           \
      acc[2 * (q) + m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(        \
          afrag[set][m][g], bfragT[n][g], acc[2 * (q) + m][n], 0, 0, 0);

  // one K-tile, phases 0..3.  Phase p: BARRIER; lgkm0; MFMA(q_p) with
  // next reads + this phase's glds interleaved under it.
#define TILE12(t)                                                           \
  do {                                                                      \
    const __bf16* At = A12((t) & 1);                                        \
    const __bf16* Atn = A12(((t) + 1) & 1);                                 \
    /* ph0 (q0 reads were issued in the previous tile's ph3 / preloop) */   \
    __builtin_amdgcn_s_barrier();                                           \
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");                      \
    __builtin_amdgcn_s_setprio(1);                                          \
    MM(0, 0);                                                               \
    RD_A(At, 1, 1);                                                         \
    if ((t) + 1 < KT) STG_A((t) + 1, 0);                                    \
    __builtin_amdgcn_s_setprio(0);                                          \
    /* ph1 */                                                               \
    __builtin_amdgcn_s_barrier();                                           \
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");                      \
    __builtin_amdgcn_s_setprio(1);                                          \
    MM(1, 1);                                                               \
    RD_A(At, 2, 0);                                                         \
    if ((t) + 1 < KT) STG_A((t) + 1, 1);                                    \
    __builtin_amdgcn_s_setprio(0);                                          \
    /* ph2: drain BEFORE B(t+2) issues (everything ph0(t+1) needs is    */  \
    /* in flight already; counted waits unsound — see v10 header)       */  \
    __builtin_amdgcn_s_barrier();                                           \
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");                      \
    __builtin_amdgcn_s_setprio(1);                                          \
    MM(2, 0);                                                               \
    RD_A(At, 3, 1);                                                         \
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");                        \
    if ((t) + 2 < KT) STG_B((t) + 2, 0);                                    \
    __builtin_amdgcn_s_setprio(0);                                          \
    /* ph3: B(t+1)/A(t+1,q0) reads issue after the last MFMA consuming  */  \
    /* bfragT/afrag — register deps order them; glds B(t+2)h1 follows   */  \
    __builtin_amdgcn_s_barrier();                                           \
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");                      \
    __builtin_amdgcn_s_setprio(1);                                          \
    MM(3, 1);                                                               \
    if ((t) + 1 < KT) {                                                     \
      RD_B(B12(((t) + 1) & 1));                                             \
      RD_A(Atn, 0, 0);                                                      \
    }                                                                       \
    if ((t) + 2 < KT) STG_B((t) + 2, 1);                                    \
    __builtin_amdgcn_s_setprio(0);                                          \
  } while (0)

  RD_B(B12(0));
  RD_A(A12(0), 0, 0);
  for (long t = 0; t < KT; ++t) TILE12(t);

  const long crow_base = bm + waveM * WM + 4 * (lane >> 4);
  const long ccol_base = bn + waveN * WN + (lane & 15);
#pragma unroll
  for (int m = 0; m < MFR; m++) {
#pragma unroll
    for (int n = 0; n < NFR; n++) {
      const long col = ccol_base + n * FRAG;
      if (col >= N) continue;
#pragma unroll
      for (int r = 0; r < 4; r++) {
        const long row = crow_base + m * FRAG + r;
        if (row >= M) continue;
        float v = acc[m][n][r];
        if (c_is_bf16)
          ((unsigned short*)C)[row * N + col] = bf16_rne(v);
        else
          ((float*)C)[row * N + col] = v;
      }
    }
  }
#undef A12
#undef B12
#undef STG_A
#undef STG_B
#undef RD_A
#undef RD_B
#undef MM
#undef TILE12
}
}  // namespace v12

// v11 = v10 with ONE barrier per phase (the trailing barrier dropped:
// the B-buffer write-after-read hazard only needs 2-phase separation,
// which the per-phase leading barrier already provides — see the
// schedule proof in cc_gemm.hip).  Sync-structure change => new race
// screen required before any production use.
extern "C" int cc_gemm_v11(const void* A, const void* B, void* C, int64_t M,
                           int64_t N, int64_t K, int c_dtype,
                           uint64_t stream) {
  if (K % (2 * BK) != 0 || K / BK < 4)
    return cc::set_error(CC_ERR_UNSUPPORTED, "K must be multiple of 128, >=256");
  dim3 block(512);
  dim3 grid((N + BN - 1) / BN, (M + BM - 1) / BM);
  hipLaunchKernelGGL((k_gemm_v10<false>), grid, block, 0, (hipStream_t)stream,
                     (const __bf16*)A, (const __bf16*)B, C, (long)M, (long)N,
                     (long)K, c_dtype == 1);
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) return cc::set_error(CC_ERR_HIP, "%s", hipGetErrorString(e));
  return CC_OK;
}

extern "C" int cc_gemm_v12(const void* A, const void* B, void* C, int64_t M,
                           int64_t N, int64_t K, int c_dtype,
                           uint64_t stream) {
  if (K % (2 * BK) != 0 || K / BK < 4)
    return cc::set_error(CC_ERR_UNSUPPORTED, "K must be multiple of 128, >=256");
  dim3 block(512);
  dim3 grid((N + BN - 1) / BN, (M + BM - 1) / BM);
  hipLaunchKernelGGL(v12::k_gemm_v12, grid, block, 0, (hipStream_t)stream,
                     (const __bf16*)A, (const __bf16*)B, C, (long)M, (long)N,
                     (long)K, c_dtype == 1);
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) return cc::set_error(CC_ERR_HIP, "%s", hipGetErrorString(e));
  return CC_OK;
}
