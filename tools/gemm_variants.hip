// GEMM structure experiments (bench-only; winner graduates into
// csrc/cc_gemm.hip).  Variants over the 128x128/BK=64 base:
//   MFMA shape: 16x16x32 (base) vs 32x32x16 (higher flops/cycle ceiling:
//               2382 vs 2075 TF µbench, MI355X_MICROARCH.md)
//   pipeline:   2-buffer vmcnt(0)+__syncthreads per K-tile (base) vs
//               3-buffer counted vmcnt + raw s_barrier (the glds-span
//               lever, cdna_hip_programming.md §5 "Pipelining across
//               barriers": +83% at 1 block/CU in the guide's microbench;
//               here LDS grows 64->96KB so occupancy drops 2->1 WG/CU —
//               measured tradeoff).
// Exported as cc_gemm_variant(variant, ...) for tools/gemm_bench2.py.

#include <hip/hip_runtime.h>

#include "../cosmos_curate_amd/csrc/cc_common.hpp"

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;

constexpr int BM = 128, BN = 128, BK = 64;

__device__ __forceinline__ unsigned short bf16_rne(float v) {
  union { float f; unsigned int u; } cv{v};
  return (unsigned short)((cv.u + 0x7fffu + ((cv.u >> 16) & 1)) >> 16);
}

__device__ __forceinline__ void stage_slice(const __bf16* __restrict__ src,
                                            long ld, long row0, long row_limit,
                                            long k0, __bf16* lds_base, int lane) {
  const int lrow8 = lane >> 3;
  const int slot = lane & 7;
  const int gk16 = slot ^ lrow8;
#pragma unroll
  for (int j = 0; j < 4; j++) {
    long grow = row0 + j * 8 + lrow8;
    grow = grow < 0 ? 0 : (grow >= row_limit ? row_limit - 1 : grow);
    const __bf16* gptr = src + grow * ld + k0 + (long)gk16 * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)gptr,
        (__attribute__((address_space(3))) unsigned int*)(lds_base + j * 8 * BK),
        16, 0, 0);
  }
}

__device__ __forceinline__ bf16x8 frag_read(const __bf16* tile, int row, int k16) {
  int slot = k16 ^ (row & 7);
  return *(const bf16x8*)(tile + (long)row * BK + slot * 8);
}

// ---- V2: 32x32x16 fragments, 2-buffer ----
// per wave 64x64 = 2x2 frags of 32x32; acc 4 x f32x16 = 64 regs.
// A frag (32x32x16): lane l holds row l&31, k 8*(l>>5).. (+8) -> 16B read.
__global__ __launch_bounds__(256, 2) void k_gemm_v2(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ C, long M, long N, long K, int c_is_bf16) {
  __shared__ __bf16 lds[2 * (BM + BN) * BK];
#define AS2(b) (lds + (b) * (BM * BK))
#define BS2(b) (lds + 2 * (BM * BK) + (b) * (BN * BK))
  const int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;
  const int waveM = wid >> 1, waveN = wid & 1;
  const long bm = (long)blockIdx.y * BM, bn = (long)blockIdx.x * BN;
  const long arow0 = bm + 32 * wid, brow0 = bn + 32 * wid;

  f32x16 acc[2][2] = {};
  const long KT = K / BK;
  stage_slice(A, K, arow0, M, 0, AS2(0) + 32 * wid * BK, lane);
  stage_slice(B, K, brow0, N, 0, BS2(0) + 32 * wid * BK, lane);
  int buf = 0;
  for (long kt = 0; kt < KT; ++kt) {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    if (kt + 1 < KT) {
      const long k0 = (kt + 1) * BK;
      stage_slice(A, K, arow0, M, k0, AS2(buf ^ 1) + 32 * wid * BK, lane);
      stage_slice(B, K, brow0, N, k0, BS2(buf ^ 1) + 32 * wid * BK, lane);
    }
    const __bf16* At = AS2(buf);
    const __bf16* Bt = BS2(buf);
    const int arow = waveM * 64 + (lane & 31);
    const int brow = waveN * 64 + (lane & 31);
#pragma unroll
    for (int kk = 0; kk < BK; kk += 16) {
      const int k16 = (kk >> 3) + (lane >> 5);  // 2 x 16B per 32 lanes
      bf16x8 a0 = frag_read(At, arow, k16);
      bf16x8 a1 = frag_read(At, arow + 32, k16);
      bf16x8 b0 = frag_read(Bt, brow, k16);
      bf16x8 b1 = frag_read(Bt, brow + 32, k16);
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b1, acc[1][1], 0, 0, 0);
    }
    buf ^= 1;
  }
  // C/D map 32x32x16: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
  const long col0 = bn + waveN * 64 + (lane & 31);
#pragma unroll
  for (int m = 0; m < 2; m++) {
#pragma unroll
    for (int n = 0; n < 2; n++) {
      const long col = col0 + n * 32;
      if (col >= N) continue;
#pragma unroll
      for (int reg = 0; reg < 16; reg++) {
        const long row = bm + waveM * 64 + m * 32 + (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
        if (row >= M) continue;
        float v = acc[m][n][reg];
        if (c_is_bf16)
          ((unsigned short*)C)[row * N + col] = bf16_rne(v);
        else
          ((float*)C)[row * N + col] = v;
      }
    }
  }
#undef AS2
#undef BS2
}

// ---- V3: 16x16x32 fragments, 3-buffer counted vmcnt + raw barrier ----
__global__ __launch_bounds__(256, 2) void k_gemm_v3(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ C, long M, long N, long K, int c_is_bf16) {
  __shared__ __bf16 lds[3 * (BM + BN) * BK];  // 96 KiB -> 1 WG/CU
#define AS3(b) (lds + (b) * (BM * BK))
#define BS3(b) (lds + 3 * (BM * BK) + (b) * (BN * BK))
  const int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;
  const int waveM = wid >> 1, waveN = wid & 1;
  const long bm = (long)blockIdx.y * BM, bn = (long)blockIdx.x * BN;
  const long arow0 = bm + 32 * wid, brow0 = bn + 32 * wid;

  f32x4 acc[4][4] = {};
  const long KT = K / BK;
  // prologue: fill 2 buffers (8 glds per buffer per wave)
  stage_slice(A, K, arow0, M, 0, AS3(0) + 32 * wid * BK, lane);
  stage_slice(B, K, brow0, N, 0, BS3(0) + 32 * wid * BK, lane);
  if (KT > 1) {
    stage_slice(A, K, arow0, M, BK, AS3(1) + 32 * wid * BK, lane);
    stage_slice(B, K, brow0, N, BK, BS3(1) + 32 * wid * BK, lane);
  }
  for (long kt = 0; kt < KT; ++kt) {
    const int buf = kt % 3;
    // wait: leave the NEXT tile's 8 glds in flight (counted vmcnt)
    if (kt + 1 < KT)
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    if (kt + 2 < KT) {
      const long k0 = (kt + 2) * BK;
      const int nb = (kt + 2) % 3;
      stage_slice(A, K, arow0, M, k0, AS3(nb) + 32 * wid * BK, lane);
      stage_slice(B, K, brow0, N, k0, BS3(nb) + 32 * wid * BK, lane);
    }
    const __bf16* At = AS3(buf);
    const __bf16* Bt = BS3(buf);
    const int arow_frag = waveM * 64 + (lane & 15);
    const int brow_frag = waveN * 64 + (lane & 15);
#pragma unroll
    for (int kk = 0; kk < BK; kk += 32) {
      const int k16 = (kk >> 3) + (lane >> 4);
      bf16x8 afrag[4], bfrag[4];
#pragma unroll
      for (int m = 0; m < 4; m++) afrag[m] = frag_read(At, arow_frag + m * 16, k16);
#pragma unroll
      for (int n = 0; n < 4; n++) bfrag[n] = frag_read(Bt, brow_frag + n * 16, k16);
#pragma unroll
      for (int m = 0; m < 4; m++)
#pragma unroll
        for (int n = 0; n < 4; n++)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag[m], bfrag[n], acc[m][n], 0, 0, 0);
    }
    // no tail barrier: the stage issued this iteration targets buf kt+2,
    // whose last readers (iteration kt-1) necessarily passed this
    // iteration's top barrier before the stage was issued.
  }
  const long crow_base = bm + waveM * 64 + 4 * (lane >> 4);
  const long ccol_base = bn + waveN * 64 + (lane & 15);
#pragma unroll
  for (int m = 0; m < 4; m++)
#pragma unroll
    for (int n = 0; n < 4; n++) {
      const long col = ccol_base + n * 16;
      if (col >= N) continue;
#pragma unroll
      for (int r = 0; r < 4; r++) {
        const long row = crow_base + m * 16 + r;
        if (row >= M) continue;
        float v = acc[m][n][r];
        if (c_is_bf16)
          ((unsigned short*)C)[row * N + col] = bf16_rne(v);
        else
          ((float*)C)[row * N + col] = v;
      }
    }
#undef AS3
#undef BS3
}

// ---- V4: 32x32x16 fragments + 3-buffer counted vmcnt ----
__global__ __launch_bounds__(256, 2) void k_gemm_v4(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ C, long M, long N, long K, int c_is_bf16) {
  __shared__ __bf16 lds[3 * (BM + BN) * BK];
#define AS4(b) (lds + (b) * (BM * BK))
#define BS4(b) (lds + 3 * (BM * BK) + (b) * (BN * BK))
  const int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;
  const int waveM = wid >> 1, waveN = wid & 1;
  const long bm = (long)blockIdx.y * BM, bn = (long)blockIdx.x * BN;
  const long arow0 = bm + 32 * wid, brow0 = bn + 32 * wid;

  f32x16 acc[2][2] = {};
  const long KT = K / BK;
  stage_slice(A, K, arow0, M, 0, AS4(0) + 32 * wid * BK, lane);
  stage_slice(B, K, brow0, N, 0, BS4(0) + 32 * wid * BK, lane);
  if (KT > 1) {
    stage_slice(A, K, arow0, M, BK, AS4(1) + 32 * wid * BK, lane);
    stage_slice(B, K, brow0, N, BK, BS4(1) + 32 * wid * BK, lane);
  }
  for (long kt = 0; kt < KT; ++kt) {
    const int buf = kt % 3;
    if (kt + 1 < KT)
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    if (kt + 2 < KT) {
      const long k0 = (kt + 2) * BK;
      const int nb = (kt + 2) % 3;
      stage_slice(A, K, arow0, M, k0, AS4(nb) + 32 * wid * BK, lane);
      stage_slice(B, K, brow0, N, k0, BS4(nb) + 32 * wid * BK, lane);
    }
    const __bf16* At = AS4(buf);
    const __bf16* Bt = BS4(buf);
    const int arow = waveM * 64 + (lane & 31);
    const int brow = waveN * 64 + (lane & 31);
#pragma unroll
    for (int kk = 0; kk < BK; kk += 16) {
      const int k16 = (kk >> 3) + (lane >> 5);
      bf16x8 a0 = frag_read(At, arow, k16);
      bf16x8 a1 = frag_read(At, arow + 32, k16);
      bf16x8 b0 = frag_read(Bt, brow, k16);
      bf16x8 b1 = frag_read(Bt, brow + 32, k16);
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b1, acc[1][1], 0, 0, 0);
    }
  }
  const long col0 = bn + waveN * 64 + (lane & 31);
#pragma unroll
  for (int m = 0; m < 2; m++)
#pragma unroll
    for (int n = 0; n < 2; n++) {
      const long col = col0 + n * 32;
      if (col >= N) continue;
#pragma unroll
      for (int reg = 0; reg < 16; reg++) {
        const long row = bm + waveM * 64 + m * 32 + (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
        if (row >= M) continue;
        float v = acc[m][n][reg];
        if (c_is_bf16)
          ((unsigned short*)C)[row * N + col] = bf16_rne(v);
        else
          ((float*)C)[row * N + col] = v;
      }
    }
#undef AS4
#undef BS4
}


// ---- V5/V6: 8-wave big tiles (256x128 / 128x256), 2-buffer ----
// Halves LDS-staged bytes per flop vs 128x128 (the L2-bandwidth bound of
// the staging loop) at unchanged waves/SIMD (512 threads, 1 WG/CU).
template <int TBM, int TBN, int WGM, int WGN>
__global__ __launch_bounds__(512, 1) void k_gemm_big(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ C, long M, long N, long K, int c_is_bf16) {
  __shared__ __bf16 lds[(TBM + TBN) * BK * 2];
  __bf16* const As0 = lds;
  __bf16* const Bs0 = lds + 2 * TBM * BK;
  const int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;  // 8 waves
  const int waveM = wid / WGN, waveN = wid % WGN;                // WGM x WGN
  const long bm = (long)blockIdx.y * TBM, bn = (long)blockIdx.x * TBN;
  // each of 8 waves stages TBM/8 rows of A and TBN/8 rows of B per tile
  const long arow0 = bm + (TBM / 8) * wid;
  const long brow0 = bn + (TBN / 8) * wid;

  f32x16 acc[2][2] = {};  // per-wave 64x64 as 2x2 of 32x32
  const long KT = K / BK;
  auto stage_rows = [&](const __bf16* src, long ld, long row0, long limit,
                        long k0, __bf16* dst, int nrows) {
    const int lrow8 = lane >> 3, slot = lane & 7, gk16 = slot ^ lrow8;
    for (int j = 0; j < nrows / 8; j++) {
      long grow = row0 + j * 8 + lrow8;
      grow = grow < 0 ? 0 : (grow >= limit ? limit - 1 : grow);
      const __bf16* gptr = src + grow * ld + k0 + (long)gk16 * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gptr,
          (__attribute__((address_space(3))) unsigned int*)(dst + j * 8 * BK),
          16, 0, 0);
    }
  };
  stage_rows(A, K, arow0, M, 0, As0 + (TBM / 8) * wid * BK, TBM / 8);
  stage_rows(B, K, brow0, N, 0, Bs0 + (TBN / 8) * wid * BK, TBN / 8);
  int buf = 0;
  for (long kt = 0; kt < KT; ++kt) {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    if (kt + 1 < KT) {
      const long k0 = (kt + 1) * BK;
      stage_rows(A, K, arow0, M, k0, As0 + (buf ^ 1) * TBM * BK + (TBM / 8) * wid * BK, TBM / 8);
      stage_rows(B, K, brow0, N, k0, Bs0 + (buf ^ 1) * TBN * BK + (TBN / 8) * wid * BK, TBN / 8);
    }
    const __bf16* At = As0 + buf * TBM * BK;
    const __bf16* Bt = Bs0 + buf * TBN * BK;
    const int arow = waveM * 64 + (lane & 31);
    const int brow = waveN * 64 + (lane & 31);
#pragma unroll
    for (int kk = 0; kk < BK; kk += 16) {
      const int k16 = (kk >> 3) + (lane >> 5);
      bf16x8 a0 = frag_read(At, arow, k16);
      bf16x8 a1 = frag_read(At, arow + 32, k16);
      bf16x8 b0 = frag_read(Bt, brow, k16);
      bf16x8 b1 = frag_read(Bt, brow + 32, k16);
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b1, acc[1][1], 0, 0, 0);
    }
    buf ^= 1;
  }
  const long col0 = bn + waveN * 64 + (lane & 31);
  const long row0 = bm + waveM * 64 + 4 * (lane >> 5);
#pragma unroll
  for (int m = 0; m < 2; m++)
#pragma unroll
    for (int n = 0; n < 2; n++) {
      const long col = col0 + n * 32;
      if (col >= N) continue;
#pragma unroll
      for (int reg = 0; reg < 16; reg++) {
        const long row = row0 + m * 32 + (reg & 3) + 8 * (reg >> 2);
        if (row >= M) continue;
        float v = acc[m][n][reg];
        if (c_is_bf16)
          ((unsigned short*)C)[row * N + col] = bf16_rne(v);
        else
          ((float*)C)[row * N + col] = v;
      }
    }
}

}  // namespace

// ---- V7: half-K ring pipeline ----
// 128x128 tile, 4-slot LDS ring of 32-deep K chunks (16 KB/slot, 64 KB
// total -> still 2 WG/CU), counted vmcnt + raw barrier per chunk: the glds
// queue never drains (prefetch distance 3 chunks).  Trades the 2-buffer
// vmcnt(0) drain for more barriers + a narrower (64 B-row) LDS image.
namespace v7 {
constexpr int CB = 32;  // chunk K depth

__device__ __forceinline__ void stage_chunk32(const __bf16* __restrict__ src,
                                              long ld, long row0, long limit,
                                              long k0, __bf16* dst, int lane) {
  // 32 rows x 32 k (64 B rows): lane l -> row l>>2, 16B slot (l&3)^((l>>2)&3)
  const int r = lane >> 2;
  const int gk16 = (lane & 3) ^ (r & 3);
#pragma unroll
  for (int j = 0; j < 2; j++) {  // 2 x 1KB = 32 rows
    long grow = row0 + j * 16 + r;
    grow = grow < 0 ? 0 : (grow >= limit ? limit - 1 : grow);
    const __bf16* g = src + grow * ld + k0 + (long)gk16 * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)g,
        (__attribute__((address_space(3))) unsigned int*)(dst + j * 16 * CB),
        16, 0, 0);
  }
}

__device__ __forceinline__ bf16x8 frag32(const __bf16* t, int row, int k16) {
  return *(const bf16x8*)(t + (long)row * CB + ((k16 ^ (row & 3)) * 8));
}

__global__ __launch_bounds__(256, 2) void k_gemm_v7(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ C, long M, long N, long K, int c_is_bf16) {
  __shared__ __bf16 lds[4 * (128 + 128) * CB];  // ring[4]: A 128x32 + B 128x32
#define A7(s) (lds + (s) * (128 * CB))
#define B7(s) (lds + 4 * (128 * CB) + (s) * (128 * CB))
  const int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;
  const int waveM = wid >> 1, waveN = wid & 1;
  const long bm = (long)blockIdx.y * 128, bn = (long)blockIdx.x * 128;
  const long arow0 = bm + 32 * wid, brow0 = bn + 32 * wid;

  f32x16 acc[2][2] = {};
  const long NC = K / CB;
  // prologue: stage chunks 0..2 (ring slots 0..2)
  for (int c = 0; c < 3 && c < NC; c++) {
    stage_chunk32(A, K, arow0, M, (long)c * CB, A7(c) + 32 * wid * CB, lane);
    stage_chunk32(B, K, brow0, N, (long)c * CB, B7(c) + 32 * wid * CB, lane);
  }
  for (long c = 0; c < NC; ++c) {
    const int slot = c & 3;
    // wait chunk c landed; leave up to 2 younger stages (8 glds) in flight
    if (c + 1 < NC)
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    if (c + 3 < NC) {  // stage chunk c+3 into slot (c+3)&3 = (c-1)&3 (free)
      const long k0 = (c + 3) * CB;
      const int ns = (c + 3) & 3;
      stage_chunk32(A, K, arow0, M, k0, A7(ns) + 32 * wid * CB, lane);
      stage_chunk32(B, K, brow0, N, k0, B7(ns) + 32 * wid * CB, lane);
    }
    const __bf16* At = A7(slot);
    const __bf16* Bt = B7(slot);
    const int ar = waveM * 64 + (lane & 31);
    const int br = waveN * 64 + (lane & 31);
#pragma unroll
    for (int kk = 0; kk < CB; kk += 16) {
      const int k16 = (kk >> 3) + (lane >> 5);
      bf16x8 a0 = frag32(At, ar, k16);
      bf16x8 a1 = frag32(At, ar + 32, k16);
      bf16x8 b0 = frag32(Bt, br, k16);
      bf16x8 b1 = frag32(Bt, br + 32, k16);
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b1, acc[1][1], 0, 0, 0);
    }
    // no tail barrier: slot (c+3)&3's last readers were chunk c-1, all past
    // this chunk's top barrier before the stage above was issued.
  }
  const long col0 = bn + waveN * 64 + (lane & 31);
  const long row0 = bm + waveM * 64 + 4 * (lane >> 5);
#pragma unroll
  for (int m = 0; m < 2; m++)
#pragma unroll
    for (int n = 0; n < 2; n++) {
      const long col = col0 + n * 32;
      if (col >= N) continue;
#pragma unroll
      for (int reg = 0; reg < 16; reg++) {
        const long row = row0 + m * 32 + (reg & 3) + 8 * (reg >> 2);
        if (row >= M) continue;
        float v = acc[m][n][reg];
        if (c_is_bf16)
          ((unsigned short*)C)[row * N + col] = bf16_rne(v);
        else
          ((float*)C)[row * N + col] = v;
      }
    }
#undef A7
#undef B7
}
}  // namespace v7


// ---- V9: 16x16x32, register-double-buffered fragments + counted waits ----
// The production body's asm shows 4x full `s_waitcnt lgkmcnt(0)` per
// K-phase: the compiler reused 24 staging VGPRs, serializing each 8-MFMA
// group on a full ds_read drain.  Here both k-groups (16 frags, 64 VGPRs)
// are read up front, ds_reads issued BEFORE the glds m0 chain, so the
// waitcnt pass can emit counted waits (lgkmcnt(8)) and the second MFMA
// group covers the remaining latency.
namespace v9 {
constexpr int WM = 64, WN = 64, FRAG = 16;
constexpr int MFR = WM / FRAG, NFR = WN / FRAG;

__global__ __launch_bounds__(256, 2) void k_gemm_v9(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ C, long M, long N, long K, int c_is_bf16) {
  __shared__ __bf16 lds[2 * (BM + BN) * BK];
#define A9(b) (lds + (b) * (BM * BK))
#define B9(b) (lds + 2 * (BM * BK) + (b) * (BN * BK))
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int waveM = wid >> 1, waveN = wid & 1;
  const long bm = (long)blockIdx.y * BM;
  const long bn = (long)blockIdx.x * BN;
  const long arow0 = bm + 32 * wid;
  const long brow0 = bn + 32 * wid;

  f32x4 acc[MFR][NFR] = {};
  const long KT = K / BK;
  stage_slice(A, K, arow0, M, 0, A9(0) + 32 * wid * BK, lane);
  stage_slice(B, K, brow0, N, 0, B9(0) + 32 * wid * BK, lane);

  const int arow_frag = waveM * WM + (lane & 15);
  const int brow_frag = waveN * WN + (lane & 15);
  int buf = 0;
  for (long kt = 0; kt < KT; ++kt) {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    const __bf16* At = A9(buf);
    const __bf16* Bt = B9(buf);
    // both k-groups up front: 16 ds_read_b128 into 64 live VGPRs
    bf16x8 afrag[2][MFR], bfrag[2][NFR];
#pragma unroll
    for (int g = 0; g < 2; g++) {
      const int k16 = (g << 2) + (lane >> 4);
#pragma unroll
      for (int m = 0; m < MFR; m++)
        afrag[g][m] = frag_read(At, arow_frag + m * FRAG, k16);
#pragma unroll
      for (int n = 0; n < NFR; n++)
        bfrag[g][n] = frag_read(Bt, brow_frag + n * FRAG, k16);
    }
    // glds for the next tile issued after the reads (vmcnt domain; the
    // m0/readfirstlane chain now runs under the ds_read latency)
    if (kt + 1 < KT) {
      const long k0 = (kt + 1) * BK;
      stage_slice(A, K, arow0, M, k0, A9(buf ^ 1) + 32 * wid * BK, lane);
      stage_slice(B, K, brow0, N, k0, B9(buf ^ 1) + 32 * wid * BK, lane);
    }
#pragma unroll
    for (int g = 0; g < 2; g++)
#pragma unroll
      for (int m = 0; m < MFR; m++)
#pragma unroll
        for (int n = 0; n < NFR; n++)
          acc[g == 0 ? m : m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[g][m], bfrag[g][n], acc[m][n], 0, 0, 0);
    buf ^= 1;
  }
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
#undef A9
#undef B9
}
}  // namespace v9

extern "C" int cc_gemm_variant(int variant, const void* A, const void* B, void* C,
                               int64_t M, int64_t N, int64_t K, int c_dtype,
                               uint64_t stream) {
  if (K % BK != 0) return cc::set_error(CC_ERR_UNSUPPORTED, "K%%64!=0");
  dim3 block(256);
  dim3 grid((N + BN - 1) / BN, (M + BM - 1) / BM);
  if (variant == 2)
    hipLaunchKernelGGL(k_gemm_v2, grid, block, 0, (hipStream_t)stream,
                       (const __bf16*)A, (const __bf16*)B, C, (long)M, (long)N,
                       (long)K, c_dtype == 1);
  else if (variant == 3)
    hipLaunchKernelGGL(k_gemm_v3, grid, block, 0, (hipStream_t)stream,
                       (const __bf16*)A, (const __bf16*)B, C, (long)M, (long)N,
                       (long)K, c_dtype == 1);
  else if (variant == 4)
    hipLaunchKernelGGL(k_gemm_v4, grid, block, 0, (hipStream_t)stream,
                       (const __bf16*)A, (const __bf16*)B, C, (long)M, (long)N,
                       (long)K, c_dtype == 1);
  else if (variant == 5) {
    dim3 g5((N + 127) / 128, (M + 255) / 256);
    hipLaunchKernelGGL((k_gemm_big<256, 128, 4, 2>), g5, dim3(512), 0,
                       (hipStream_t)stream, (const __bf16*)A, (const __bf16*)B,
                       C, (long)M, (long)N, (long)K, c_dtype == 1);
  } else if (variant == 6) {
    dim3 g6((N + 255) / 256, (M + 127) / 128);
    hipLaunchKernelGGL((k_gemm_big<128, 256, 2, 4>), g6, dim3(512), 0,
                       (hipStream_t)stream, (const __bf16*)A, (const __bf16*)B,
                       C, (long)M, (long)N, (long)K, c_dtype == 1);
  } else if (variant == 7) {
    if (K % 32 != 0) return cc::set_error(CC_ERR_UNSUPPORTED, "K%%32");
    hipLaunchKernelGGL(v7::k_gemm_v7, grid, block, 0, (hipStream_t)stream,
                       (const __bf16*)A, (const __bf16*)B, C, (long)M,
                       (long)N, (long)K, c_dtype == 1);
  } else if (variant == 9) {
    hipLaunchKernelGGL(v9::k_gemm_v9, grid, block, 0, (hipStream_t)stream,
                       (const __bf16*)A, (const __bf16*)B, C, (long)M,
                       (long)N, (long)K, c_dtype == 1);
  } else
    return cc::set_error(CC_ERR_INVALID, "variant must be 2|3|4|5|6|7|9");
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) return cc::set_error(CC_ERR_HIP, "%s", hipGetErrorString(e));
  return CC_OK;
}
