// V8: deep-pipelined 256x256 GEMM — derived phase schedule (not the
// guide's template verbatim; same levers: big tile, per-phase staging,
// counted vmcnt, raw barriers, setprio).
//
// Geometry: 512 threads = 8 waves (2M x 4N), per-wave 128x64 output as
// 4 quadrant phases of 32x64 (2 x mfma_f32_32x32x16_bf16 per kstep16).
// LDS 160 KiB: A = ring of 8 phase-slots (8 KiB: both M-halves' 32-row
// quadrant x 64k), B = ring of 3 tile-buffers (32 KiB: 256 rows x 64k).
// Steady state: every phase issues exactly 2 glds (1 A slot for phase
// p+4, 1/4 of B tile T+2) and waits vmcnt(6) (= the 3 younger phases'
// issues) before one raw barrier -> all staged data consumed this phase
// landed chip-wide.  Schedule proof sketch:
//   A slot s=p%8: issued p-4, landed by p (vmcnt), read at p, re-issued
//     p+4 (after read) -> no overlap.
//   B buf T%3: issued 4(T-2)..+3 (distance 5..8), read 4T..4T+3,
//     re-issued 4(T+3-2)=4T+4.. (after last read).
#include <hip/hip_runtime.h>
#include "../cosmos_curate_amd/csrc/cc_common.hpp"

namespace v8 {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

constexpr int BM = 256, BN = 256, BK = 64;

__device__ __forceinline__ unsigned short bf16_rne(float v) {
  union { float f; unsigned int u; } cv{v};
  return (unsigned short)((cv.u + 0x7fffu + ((cv.u >> 16) & 1)) >> 16);
}

// ---- staging ----------------------------------------------------------
// A phase-slot: 64 rows (2 half-quarters) x 64 k, one glds of 512 lanes.
// thread t: slot row = t>>3 (0..63), 16B chunk (t&7), source swizzle
// gk16 = (t&7) ^ (row & 7).
__device__ __forceinline__ void stage_A_slot(const __bf16* __restrict__ A,
                                             long K, long M, long bm, int q,
                                             __bf16* slot, int tid) {
  const int lrow = tid >> 3;
  const int gk16 = (tid & 7) ^ (lrow & 7);
  // global row: halves 0/1 of the block's 256 rows, quadrant q
  long grow = bm + (lrow < 32 ? 32 * q + lrow : 128 + 32 * q + (lrow - 32));
  grow = grow < 0 ? 0 : (grow >= M ? M - 1 : grow);
  const long k0 = 0;  // caller adds tile k offset into A pointer
  const __bf16* g = A + grow * K + k0 + (long)gk16 * 8;
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)g,
      (__attribute__((address_space(3))) unsigned int*)(slot + (long)lrow * BK),
      16, 0, 0);
}

// B tile staged as 4 chunks of 64 rows; chunk c: rows 64c..64c+63.
__device__ __forceinline__ void stage_B_chunk(const __bf16* __restrict__ B,
                                              long K, long N, long bn, int c,
                                              __bf16* buf, int tid) {
  const int lrow = tid >> 3;        // 0..63 within chunk
  const int gk16 = (tid & 7) ^ (lrow & 7);
  long grow = bn + 64 * c + lrow;
  grow = grow < 0 ? 0 : (grow >= N ? N - 1 : grow);
  const __bf16* g = B + grow * K + (long)gk16 * 8;
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)g,
      (__attribute__((address_space(3))) unsigned int*)(buf + (long)(64 * c + lrow) * BK),
      16, 0, 0);
}

__device__ __forceinline__ bf16x8 fragA(const __bf16* slot, int half,
                                        int lane, int k16) {
  const int row = half * 32 + (lane & 31);
  return *(const bf16x8*)(slot + (long)row * BK + ((k16 ^ (row & 7)) * 8));
}

__device__ __forceinline__ bf16x8 fragB(const __bf16* buf, int brow, int k16) {
  return *(const bf16x8*)(buf + (long)brow * BK + ((k16 ^ (brow & 7)) * 8));
}

__global__ __launch_bounds__(512, 1) void k_gemm_v8(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ C, long M, long N, long K, int c_is_bf16) {
  __shared__ __bf16 lds[(8 * 64 + 3 * 256) * BK];  // A ring 64K + B ring 96K
#define ASLOT(s) (lds + (long)(s) * (64 * BK))
#define BBUF(b) (lds + 8L * 64 * BK + (long)(b) * (256 * BK))
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;              // 8 waves
  const int waveM = wid >> 2, waveN = wid & 3;
  const long bm = (long)blockIdx.y * BM, bn = (long)blockIdx.x * BN;

  f32x16 acc[4][2] = {};  // [quadrant][col-half-of-64]

  const long KT = K / BK;
  // ---- prologue: B tiles 0,1 + A slots for phases 0..3 (tile 0) -------
  for (int c = 0; c < 4; c++) stage_B_chunk(B, K, N, bn, c, BBUF(0), tid);
  for (int q = 0; q < 4; q++)
    stage_A_slot(A + 0, K, M, bm, q, ASLOT(q), tid);
  if (KT > 1)
    for (int c = 0; c < 4; c++)
      stage_B_chunk(B + BK, K, N, bn, c, BBUF(1), tid);
  // outstanding: 12 glds (B0 x4, A0..A3, B1 x4)

  for (long kt = 0; kt < KT; ++kt) {
    const __bf16* bb = BBUF(kt % 3);
#pragma unroll
    for (int q = 0; q < 4; q++) {
      const long p = 4 * kt + q;
      // wait: everything issued strictly more than 3 phases ago must land.
      // steady state: 2 glds/phase -> vmcnt(6); the first 4 phases drain
      // the prologue shape instead (outstanding after phase-p issues is
      // kept <= 8; constants below derived from the issue sequence).
      if (kt == 0) {
        // prologue issue order: B0 x4, A0..A3, B1 x4 (12 glds).  Phase q
        // needs B0 + A-slot q landed; the allowed-outstanding count is
        // "everything younger than A_q": (3-q skipped A's) + B1's 4 +
        // 2 glds issued by each earlier phase of this tile.
        if (q == 0)
          asm volatile("s_waitcnt vmcnt(7)" ::: "memory");
        else if (q == 1)
          asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
        else if (q == 2)
          asm volatile("s_waitcnt vmcnt(9)" ::: "memory");
        else
          asm volatile("s_waitcnt vmcnt(10)" ::: "memory");
      } else if (kt == KT - 2) {
        // B staging stopped (kt+2 == KT): issue rate 1/phase; the wait
        // must equal the glds issued in the last 3 phases exactly.
        if (q == 0)      asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
        else if (q == 1) asm volatile("s_waitcnt vmcnt(5)" ::: "memory");
        else if (q == 2) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        else             asm volatile("s_waitcnt vmcnt(3)" ::: "memory");
      } else if (kt == KT - 1) {
        // A staging stopped too: 0/phase
        if (q == 0)      asm volatile("s_waitcnt vmcnt(3)" ::: "memory");
        else if (q == 1) asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
        else if (q == 2) asm volatile("s_waitcnt vmcnt(1)" ::: "memory");
        else             asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      } else {
        asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
      }
      __builtin_amdgcn_s_barrier();

      // ---- issue this phase's staging (future work) ----
      // A slot for phase p+4 (same tile kt+1's quadrant q, or tile kt+1)
      if (p + 4 < 4 * KT) {
        const long kt_a = (p + 4) >> 2;
        const int q_a = (p + 4) & 3;
        stage_A_slot(A + kt_a * BK, K, M, bm, q_a, ASLOT((p + 4) & 7), tid);
      }
      // B chunk q of tile kt+2
      if (kt + 2 < KT)
        stage_B_chunk(B + (kt + 2) * BK, K, N, bn, q, BBUF((kt + 2) % 3), tid);

      // ---- compute quadrant q over K=64 ----
      const __bf16* aslot = ASLOT(p & 7);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < BK; kk += 16) {
        const int k16 = (kk >> 3) + (lane >> 5);
        bf16x8 af = fragA(aslot, waveM, lane, k16);
        bf16x8 b0 = fragB(bb, waveN * 64 + (lane & 31), k16);
        bf16x8 b1 = fragB(bb, waveN * 64 + 32 + (lane & 31), k16);
        acc[q][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, b0, acc[q][0], 0, 0, 0);
        acc[q][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, b1, acc[q][1], 0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);
    }
  }

  // ---- epilogue: quadrant q rows bm + waveM*128 + 32q + C/D 32x32 map --
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  const long col0 = bn + waveN * 64 + (lane & 31);
  const bool interior = (bm + BM <= M) && (bn + BN <= N);
#pragma unroll
  for (int q = 0; q < 4; q++) {
#pragma unroll
    for (int nh = 0; nh < 2; nh++) {
      const long col = col0 + nh * 32;
      if (!interior && col >= N) continue;
#pragma unroll
      for (int reg = 0; reg < 16; reg++) {
        const long row = bm + waveM * 128 + 32 * q + (reg & 3) + 8 * (reg >> 2)
                         + 4 * (lane >> 5);
        if (!interior && row >= M) continue;
        float v = acc[q][nh][reg];
        if (c_is_bf16)
          ((unsigned short*)C)[row * N + col] = bf16_rne(v);
        else
          ((float*)C)[row * N + col] = v;
      }
    }
  }
#undef ASLOT
#undef BBUF
}

}  // namespace v8

extern "C" int cc_gemm_v8(const void* A, const void* B, void* C, int64_t M,
                          int64_t N, int64_t K, int c_dtype, uint64_t stream) {
  if (K % 64 != 0 || K / 64 < 5)
    return cc::set_error(CC_ERR_UNSUPPORTED, "v8 needs K%%64==0 and K>=320");
  dim3 block(512);
  dim3 grid((N + 255) / 256, (M + 255) / 256);
  hipLaunchKernelGGL(v8::k_gemm_v8, grid, block, 0, (hipStream_t)stream,
                     (const __bf16*)A, (const __bf16*)B, C, (long)M, (long)N,
                     (long)K, c_dtype == 1);
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) return cc::set_error(CC_ERR_HIP, "%s", hipGetErrorString(e));
  return CC_OK;
}
