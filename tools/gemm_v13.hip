// V13/V14: register-staged 256x256 GEMM pipeline (round-2 experiment).
//
// Why: the production t256 body (csrc/cc_gemm.hip k_gemm_bf16_t256)
// carries one full `s_waitcnt vmcnt(0)` drain per K-tile because counted
// vmcnt waits on global_load_lds were measured NONDETERMINISTIC on this
// pool (profiles/r01_t256_det2.log — LDS-DMA completions retire out of
// order under load).  That drain exposes raw HBM latency once per K-tile
// with 1 WG/CU and nothing to cover it; at the short-K ViT shapes
// (KT=12) it plus prologue/epilogue is most of the gap to the >=40%
// MFMA target (BENCH_r01 frac 0.29).
//
// Fix tried here: stage through REGISTERS — plain global loads into
// VGPRs, ds_write into LDS.  Plain loads are scoreboarded per register
// by hipcc (in-order vmcnt semantics hold for VMEM-to-VGPR returns), so
// the compiler inserts precise counted waits and the K-loop carries NO
// global drain at all.  Schedule per K-tile t (buf = t&1, 4 phases, one
// leading barrier per phase — the v11 discipline, hazards separated by
// >=1 barrier via the pre-barrier-issue property):
//   ph1: ds_read B(t) all + A q0; ds_write A(t+1) (4 dsw -> buf^1)
//   ph2: ds_read A q1;            ds_write B(t+1) (4 dsw);
//        global loads A(t+2) -> aregs (4)
//   ph3: ds_read A q2;            global loads B(t+2) -> bregs (4)
//   ph4: ds_read A q3
// Hazard check (all 1+ barrier separated by pre-barrier issue):
//   WAR buf^1: tile t-1's last reads of buf^1 issue at its ph4
//     pre-barrier; writes issue at tile t ph1 pre-barrier (1 barrier).
//   RAW next tile: writes done by ph2 pre-barrier; next-tile reads
//     issue at ph4 pre-barrier / t+1 ph1 (2 barriers).
//   Register WAR: aregs consumed by dsw at ph1, reloaded ph2; bregs
//     consumed ph2, reloaded ph3 (compiler-enforced anyway).
// V14 = V13 + persistent multi-tile outer loop: one WG computes several
// output tiles; the next tile's global loads issue BEFORE the epilogue
// stores, so the prologue latency hides under the C writeback.
//
// Verification: sync structure is NEW -> two-lane discipline: refcheck
// vs production at small sizes + multi-run determinism screen at the
// big ViT grids (tools/gemm_v13_screen.py).

#include <hip/hip_runtime.h>

#include "../cosmos_curate_amd/csrc/cc_common.hpp"

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

constexpr int BM = 256, BN = 256, BK = 64;
constexpr int WM = 128, WN = 64;  // per-wave output (2M x 4N wave grid)
constexpr int FRAG = 16;
constexpr int MFR = WM / FRAG;  // 8
constexpr int NFR = WN / FRAG;  // 4

__device__ __forceinline__ unsigned short bf16_rne(float v) {
  union { float f; unsigned int u; } cv{v};
  return (unsigned short)((cv.u + 0x7fffu + ((cv.u >> 16) & 1)) >> 16);
}

__device__ __forceinline__ bf16x8 frag_read(const __bf16* tile, int row,
                                            int k16) {
  int slot = k16 ^ (row & 7);
  return *(const bf16x8*)(tile + (long)row * BK + slot * 8);
}

// ---- register staging ----
// A wave owns 32 rows of each 256-row operand tile per K-tile, as 4
// groups of 8 rows: groups 0,1 = half0 rows wid*16..+16, groups 2,3 =
// half1 rows 128+wid*16..+16.  Lane covers (row = grp + (lane>>3),
// slot = lane&7): one bf16x8 (16 B) per group -> coalesced 128 B per
// 8 lanes, identical addressing to the glds path's source.
__device__ __forceinline__ long stage_row(int wid, int lane, int g) {
  return (long)((g >> 1) * 128 + wid * 16 + (g & 1) * 8 + (lane >> 3));
}

#define LOAD_REGS(dst, src, ld, base, limit, k0)                            \
  _Pragma("unroll") for (int g = 0; g < 4; g++) {                           \
    long grow = (base) + stage_row(wid, lane, g);                           \
    grow = grow < 0 ? 0 : (grow >= (limit) ? (limit)-1 : grow);             \
    dst[g] = *(const bf16x8*)((src) + grow * (ld) + (k0) + slot * 8);       \
  }

#define WRITE_REGS(srcreg, ldsbase)                                         \
  _Pragma("unroll") for (int g = 0; g < 4; g++) {                           \
    long lrow = stage_row(wid, lane, g);                                    \
    *(bf16x8*)((ldsbase) + lrow * BK + ((long)(slot ^ (lrow & 7))) * 8) =   \
        srcreg[g];                                                          \
  }

template <int PERSIST>  // 0: one tile per WG; 1: grid-strided tiles
__global__ __launch_bounds__(512, 1) void k_gemm_v13(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ C, long M, long N, long K, int c_is_bf16, int nbx,
    int nwg, int do_remap) {
  __shared__ __bf16 lds[2 * (BM + BN) * BK];  // 128 KiB
#define A13(b) (lds + (b) * (BM * BK))
#define B13(b) (lds + 2 * (BM * BK) + (b) * (BN * BK))
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int waveM = wid >> 2, waveN = wid & 3;
  const int slot = lane & 7;
  const long KT = K / BK;  // caller guarantees KT >= 2

  const int arow_base = waveM * WM + (lane & 15);
  const int brow_base = waveN * WN + (lane & 15);

  bf16x8 aregs[4], bregs[4];
  bf16x8 bfragT[NFR][2];
  bf16x8 afrag[2][2];

  const int tiles = PERSIST ? (nwg + (int)gridDim.x - 1) / (int)gridDim.x : 1;
  for (int rep = 0; rep < tiles; rep++) {
    int orig = PERSIST ? (int)blockIdx.x + rep * (int)gridDim.x
                       : (int)blockIdx.x;
    if (orig >= nwg) break;
    if (do_remap) {
      int q = nwg >> 3, r = nwg & 7;
      int xcd = orig & 7, lid = orig >> 3;
      orig = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + lid;
    }
    const long bm = (long)(orig / nbx) * BM;
    const long bn = (long)(orig % nbx) * BN;

    f32x4 acc[MFR][NFR] = {};

    // prologue: tile 0 via regs (on rep>0 the loads were issued before
    // the previous epilogue, hiding their latency under the C stores)
    if (rep == 0) {
      LOAD_REGS(aregs, A, K, bm, M, 0);
      LOAD_REGS(bregs, B, K, bn, N, 0);
    }
    WRITE_REGS(aregs, A13(0));
    WRITE_REGS(bregs, B13(0));
    if (KT > 1) {
      LOAD_REGS(aregs, A, K, bm, M, BK);
      LOAD_REGS(bregs, B, K, bn, N, BK);
    }
    __syncthreads();  // publish buf0 (drains the ds_writes)

#define PHASE_MFMA13(q)                                                     \
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
  } while (0)

#define READ_A13(At, q)                                                     \
  _Pragma("unroll") for (int g = 0; g < 2; g++) {                           \
    const int k16 = (g << 2) + (lane >> 4);                                 \
    _Pragma("unroll") for (int m = 0; m < 2; m++) afrag[m][g] =             \
        frag_read(At, arow_base + (q) * 32 + m * FRAG, k16);                \
  }

#define READ_B13(Bt)                                                        \
  _Pragma("unroll") for (int g = 0; g < 2; g++) {                           \
    const int k16 = (g << 2) + (lane >> 4);                                 \
    _Pragma("unroll") for (int n = 0; n < NFR; n++) bfragT[n][g] =          \
        frag_read(Bt, brow_base + n * FRAG, k16);                           \
  }

    for (long t = 0; t < KT; ++t) {
      const __bf16* At = A13(t & 1);
      const __bf16* Bt = B13(t & 1);
      __bf16* Aw = A13((t & 1) ^ 1);
      __bf16* Bw = B13((t & 1) ^ 1);
      const bool stage_next = t + 1 < KT;    // write t+1 into buf^1
      const bool load_next2 = t + 2 < KT;    // load t+2 into regs
      // ph1
      READ_B13(Bt);
      READ_A13(At, 0);
      if (stage_next) WRITE_REGS(aregs, Aw);
      __builtin_amdgcn_s_barrier();
      PHASE_MFMA13(0);
      // ph2
      READ_A13(At, 1);
      if (stage_next) WRITE_REGS(bregs, Bw);
      if (load_next2) LOAD_REGS(aregs, A, K, bm, M, (t + 2) * BK);
      __builtin_amdgcn_s_barrier();
      PHASE_MFMA13(1);
      // ph3
      READ_A13(At, 2);
      if (load_next2) LOAD_REGS(bregs, B, K, bn, N, (t + 2) * BK);
      __builtin_amdgcn_s_barrier();
      PHASE_MFMA13(2);
      // ph4
      READ_A13(At, 3);
      __builtin_amdgcn_s_barrier();
      PHASE_MFMA13(3);
    }

    // persistent: issue the NEXT output tile's prologue loads before the
    // epilogue stores so their HBM latency hides under the writeback.
    if (PERSIST && rep + 1 < tiles) {
      int nxt = (int)blockIdx.x + (rep + 1) * (int)gridDim.x;
      if (nxt < nwg) {
        int o2 = nxt;
        if (do_remap) {
          int q = nwg >> 3, r = nwg & 7;
          int xcd = o2 & 7, lid = o2 >> 3;
          o2 = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + lid;
        }
        const long bm2 = (long)(o2 / nbx) * BM;
        const long bn2 = (long)(o2 % nbx) * BN;
        LOAD_REGS(aregs, A, K, bm2, M, 0);
        LOAD_REGS(bregs, B, K, bn2, N, 0);
      }
    }

    // epilogue (same C/D map as production)
    const long crow_base = bm + waveM * WM + 4 * (lane >> 4);
    const long ccol_base = bn + waveN * WN + (lane & 15);
    const bool interior = (bm + BM <= M) && (bn + BN <= N);
    if (interior) {
#pragma unroll
      for (int m = 0; m < MFR; m++) {
#pragma unroll
        for (int n = 0; n < NFR; n++) {
          const long col = ccol_base + n * FRAG;
#pragma unroll
          for (int r = 0; r < 4; r++) {
            const long row = crow_base + m * FRAG + r;
            if (c_is_bf16)
              ((unsigned short*)C)[row * N + col] = bf16_rne(acc[m][n][r]);
            else
              ((float*)C)[row * N + col] = acc[m][n][r];
          }
        }
      }
    } else {
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
            if (c_is_bf16)
              ((unsigned short*)C)[row * N + col] = bf16_rne(acc[m][n][r]);
            else
              ((float*)C)[row * N + col] = acc[m][n][r];
          }
        }
      }
    }
    if (PERSIST && rep + 1 < tiles) __syncthreads();  // LDS reuse fence
  }
#undef A13
#undef B13
#undef PHASE_MFMA13
#undef READ_A13
#undef READ_B13
}

}  // namespace

// persist: 0 = one tile per WG (grid = nwg), 1 = persistent (grid =
// min(nwg, 8*wgs_per_xcd*... caller passes grid via the env knob below).
extern "C" int cc_gemm_v13(const void* A, const void* B, void* C, long M,
                           long N, long K, int c_is_bf16, int persist,
                           int remap, unsigned long long stream) {
  if (K % BK != 0 || K / BK < 2) return -2;
  int nbx = (int)((N + BN - 1) / BN);
  int nby = (int)((M + BM - 1) / BM);
  int nwg = nbx * nby;
  int grid = nwg;
  if (persist) grid = nwg < 256 ? nwg : 256;  // 1 WG/CU persistent fleet
  if (persist)
    hipLaunchKernelGGL((k_gemm_v13<1>), dim3(grid), dim3(512), 0,
                       (hipStream_t)stream, (const __bf16*)A,
                       (const __bf16*)B, C, M, N, K, c_is_bf16, nbx, nwg,
                       remap);
  else
    hipLaunchKernelGGL((k_gemm_v13<0>), dim3(grid), dim3(512), 0,
                       (hipStream_t)stream, (const __bf16*)A,
                       (const __bf16*)B, C, M, N, K, c_is_bf16, nbx, nwg,
                       remap);
  return hipGetLastError() == hipSuccess ? 0 : -1;
}

// ---- V15: production glds 8-phase schedule + persistent multi-tile ----
// v13 showed register staging costs more than the glds path's vmcnt(0)
// drain (704 vs 791 at patch64), but persistence + remap wins at the
// skinny-N ViT shapes (fc1 784 vs 705).  v15 keeps the production
// KTILE2 glds schedule verbatim and adds the persistent outer loop,
// issuing the NEXT tile's 12-glds prologue BEFORE the epilogue stores
// so its HBM latency hides under the C writeback.
namespace {

__device__ __forceinline__ void stage_half15(const __bf16* __restrict__ src,
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

template <int PERSIST>
__global__ __launch_bounds__(512, 1) void k_gemm_v15(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ C, long M, long N, long K, int c_is_bf16, int nbx,
    int nwg, int do_remap) {
  __shared__ __bf16 lds[2 * (BM + BN) * BK];  // 128 KiB
#define A15(b) (lds + (b) * (BM * BK))
#define B15(b) (lds + 2 * (BM * BK) + (b) * (BN * BK))
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int waveM = wid >> 2, waveN = wid & 3;
  const long KT = K / BK;  // caller guarantees even, >= 4
  const long srow = wid * 16;
  const int arow_base = waveM * WM + (lane & 15);
  const int brow_base = waveN * WN + (lane & 15);
  bf16x8 bfragT[NFR][2];
  bf16x8 afrag[2][2];

#define STAGE_A15(bmv, t, h)                                                \
  stage_half15(A, K, (bmv) + (h) * 128 + srow, M, (t) * BK,                 \
               A15((t) & 1) + ((h) * 128 + srow) * BK, lane)
#define STAGE_B15(bnv, t, h)                                                \
  stage_half15(B, K, (bnv) + (h) * 128 + srow, N, (t) * BK,                 \
               B15((t) & 1) + ((h) * 128 + srow) * BK, lane)

  // tile index for rep r of this WG under the three order modes:
  //   0: orig = bid + r*grid                  (stride)
  //   1: xcd-remap(orig) of mode 0            (stride + L2 affinity)
  //   2: orig = chunk_start(bid) + r          (contiguous row-major chunk:
  //      the A row band stays hot in L2 across ~nbx consecutive reps)
  int chunk_start = 0, chunk_len = 1;
  if (PERSIST && do_remap == 2) {
    int q = nwg / (int)gridDim.x, r = nwg % (int)gridDim.x;
    int b = (int)blockIdx.x;
    chunk_len = q + (b < r ? 1 : 0);
    chunk_start = b * q + (b < r ? b : r);
  }
  const int tiles =
      PERSIST ? (do_remap == 2 ? chunk_len
                               : (nwg + (int)gridDim.x - 1) / (int)gridDim.x)
              : 1;
  long bm = 0, bn = 0;
  {
    int orig = do_remap == 2 ? chunk_start : (int)blockIdx.x;
    if (do_remap == 1) {
      int q = nwg >> 3, r = nwg & 7;
      int xcd = orig & 7, lid = orig >> 3;
      orig = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + lid;
    }
    bm = (long)(orig / nbx) * BM;
    bn = (long)(orig % nbx) * BN;
  }
  // first prologue (full-latency; later ones hide under epilogues)
  STAGE_A15(bm, 0, 0);
  STAGE_A15(bm, 0, 1);
  STAGE_B15(bn, 0, 0);
  STAGE_B15(bn, 0, 1);
  STAGE_B15(bn, 1, 0);
  STAGE_B15(bn, 1, 1);

  for (int rep = 0; rep < tiles; rep++) {
    if (PERSIST && do_remap != 2 &&
        (int)blockIdx.x + rep * (int)gridDim.x >= nwg)
      break;
    f32x4 acc[MFR][NFR] = {};
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

#define PHASE_MFMA15(q)                                                     \
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
  } while (0)

#define READ_A15(At, q)                                                     \
  _Pragma("unroll") for (int g = 0; g < 2; g++) {                           \
    const int k16 = (g << 2) + (lane >> 4);                                 \
    _Pragma("unroll") for (int m = 0; m < 2; m++) afrag[m][g] =             \
        frag_read(At, arow_base + (q) * 32 + m * FRAG, k16);                \
  }

#define READ_B15(Bt)                                                        \
  _Pragma("unroll") for (int g = 0; g < 2; g++) {                           \
    const int k16 = (g << 2) + (lane >> 4);                                 \
    _Pragma("unroll") for (int n = 0; n < NFR; n++) bfragT[n][g] =          \
        frag_read(Bt, brow_base + n * FRAG, k16);                           \
  }

#define KTILE15(t)                                                          \
  do {                                                                      \
    const __bf16* At = A15((t) & 1);                                        \
    const __bf16* Bt = B15((t) & 1);                                        \
    READ_B15(Bt);                                                           \
    READ_A15(At, 0);                                                        \
    if ((t) + 1 < KT) STAGE_A15(bm, (t) + 1, 0);                            \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA15(0);                                                        \
    READ_A15(At, 1);                                                        \
    if ((t) + 1 < KT) STAGE_A15(bm, (t) + 1, 1);                            \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA15(1);                                                        \
    READ_A15(At, 2);                                                        \
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");                        \
    if ((t) + 2 < KT) STAGE_B15(bn, (t) + 2, 0);                            \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA15(2);                                                        \
    READ_A15(At, 3);                                                        \
    if ((t) + 2 < KT) STAGE_B15(bn, (t) + 2, 1);                            \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA15(3);                                                        \
  } while (0)

    for (long it = 0; it < KT / 2; ++it) {
      KTILE15(2 * it);
      KTILE15(2 * it + 1);
    }

    const long ebm = bm, ebn = bn;
    // advance to next tile and issue its prologue BEFORE the epilogue:
    // the LDS buffers are fully consumed (every wave drained its reads
    // via lgkmcnt(0) before its last barrier), but another wave may
    // still be inside the last phase -> barrier first.
    if (PERSIST && rep + 1 < tiles &&
        (do_remap == 2 ||
         (int)blockIdx.x + (rep + 1) * (int)gridDim.x < nwg)) {
      __builtin_amdgcn_s_barrier();
      int orig = do_remap == 2
                     ? chunk_start + rep + 1
                     : (int)blockIdx.x + (rep + 1) * (int)gridDim.x;
      if (do_remap == 1) {
        int q = nwg >> 3, r = nwg & 7;
        int xcd = orig & 7, lid = orig >> 3;
        orig = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + lid;
      }
      bm = (long)(orig / nbx) * BM;
      bn = (long)(orig % nbx) * BN;
      STAGE_A15(bm, 0, 0);
      STAGE_A15(bm, 0, 1);
      STAGE_B15(bn, 0, 0);
      STAGE_B15(bn, 0, 1);
      STAGE_B15(bn, 1, 0);
      STAGE_B15(bn, 1, 1);
    }

    const long crow_base = ebm + waveM * WM + 4 * (lane >> 4);
    const long ccol_base = ebn + waveN * WN + (lane & 15);
    const bool interior = (ebm + BM <= M) && (ebn + BN <= N);
    if (interior) {
#pragma unroll
      for (int m = 0; m < MFR; m++) {
#pragma unroll
        for (int n = 0; n < NFR; n++) {
          const long col = ccol_base + n * FRAG;
#pragma unroll
          for (int r = 0; r < 4; r++) {
            const long row = crow_base + m * FRAG + r;
            if (c_is_bf16)
              ((unsigned short*)C)[row * N + col] = bf16_rne(acc[m][n][r]);
            else
              ((float*)C)[row * N + col] = acc[m][n][r];
          }
        }
      }
    } else {
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
            if (c_is_bf16)
              ((unsigned short*)C)[row * N + col] = bf16_rne(acc[m][n][r]);
            else
              ((float*)C)[row * N + col] = acc[m][n][r];
          }
        }
      }
    }
  }
#undef A15
#undef B15
#undef STAGE_A15
#undef STAGE_B15
#undef PHASE_MFMA15
#undef READ_A15
#undef READ_B15
#undef KTILE15
}

}  // namespace

extern "C" int cc_gemm_v15(const void* A, const void* B, void* C, long M,
                           long N, long K, int c_is_bf16, int persist,
                           int remap, unsigned long long stream) {
  if (K % (2 * BK) != 0 || K / BK < 4) return -2;
  int nbx = (int)((N + BN - 1) / BN);
  int nby = (int)((M + BM - 1) / BM);
  int nwg = nbx * nby;
  int grid = nwg;
  if (persist) grid = nwg < 256 ? nwg : 256;
  if (persist)
    hipLaunchKernelGGL((k_gemm_v15<1>), dim3(grid), dim3(512), 0,
                       (hipStream_t)stream, (const __bf16*)A,
                       (const __bf16*)B, C, M, N, K, c_is_bf16, nbx, nwg,
                       remap);
  else
    hipLaunchKernelGGL((k_gemm_v15<0>), dim3(grid), dim3(512), 0,
                       (hipStream_t)stream, (const __bf16*)A,
                       (const __bf16*)B, C, M, N, K, c_is_bf16, nbx, nwg,
                       remap);
  return hipGetLastError() == hipSuccess ? 0 : -1;
}

// ---- V19: v15 with counted vmcnt publish (drain removed) ----
// v13 showed register staging costs more than the glds path's vmcnt(0)
// drain (704 vs 791 at patch64), but persistence + remap wins at the
// skinny-N ViT shapes (fc1 784 vs 705).  v15 keeps the production
// KTILE2 glds schedule verbatim and adds the persistent outer loop,
// issuing the NEXT tile's 12-glds prologue BEFORE the epilogue stores
// so its HBM latency hides under the C writeback.
namespace {

template <int PERSIST>
__global__ __launch_bounds__(512, 1) void k_gemm_v19(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ C, long M, long N, long K, int c_is_bf16, int nbx,
    int nwg, int do_remap) {
  __shared__ __bf16 lds[2 * (BM + BN) * BK];  // 128 KiB
#define A19(b) (lds + (b) * (BM * BK))
#define B19(b) (lds + 2 * (BM * BK) + (b) * (BN * BK))
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int waveM = wid >> 2, waveN = wid & 3;
  const long KT = K / BK;  // caller guarantees even, >= 4
  const long srow = wid * 16;
  const int arow_base = waveM * WM + (lane & 15);
  const int brow_base = waveN * WN + (lane & 15);
  bf16x8 bfragT[NFR][2];
  bf16x8 afrag[2][2];

#define STAGE_A19(bmv, t, h)                                                \
  stage_half15(A, K, (bmv) + (h) * 128 + srow, M, (t) * BK,                 \
               A19((t) & 1) + ((h) * 128 + srow) * BK, lane)
#define STAGE_B19(bnv, t, h)                                                \
  stage_half15(B, K, (bnv) + (h) * 128 + srow, N, (t) * BK,                 \
               B19((t) & 1) + ((h) * 128 + srow) * BK, lane)

  // tile index for rep r of this WG under the three order modes:
  //   0: orig = bid + r*grid                  (stride)
  //   1: xcd-remap(orig) of mode 0            (stride + L2 affinity)
  //   2: orig = chunk_start(bid) + r          (contiguous row-major chunk:
  //      the A row band stays hot in L2 across ~nbx consecutive reps)
  int chunk_start = 0, chunk_len = 1;
  if (PERSIST && do_remap == 2) {
    int q = nwg / (int)gridDim.x, r = nwg % (int)gridDim.x;
    int b = (int)blockIdx.x;
    chunk_len = q + (b < r ? 1 : 0);
    chunk_start = b * q + (b < r ? b : r);
  }
  const int tiles =
      PERSIST ? (do_remap == 2 ? chunk_len
                               : (nwg + (int)gridDim.x - 1) / (int)gridDim.x)
              : 1;
  long bm = 0, bn = 0;
  {
    int orig = do_remap == 2 ? chunk_start : (int)blockIdx.x;
    if (do_remap == 1) {
      int q = nwg >> 3, r = nwg & 7;
      int xcd = orig & 7, lid = orig >> 3;
      orig = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + lid;
    }
    bm = (long)(orig / nbx) * BM;
    bn = (long)(orig % nbx) * BN;
  }
  // first prologue (full-latency; later ones hide under epilogues)
  STAGE_A19(bm, 0, 0);
  STAGE_A19(bm, 0, 1);
  STAGE_B19(bn, 0, 0);
  STAGE_B19(bn, 0, 1);
  STAGE_B19(bn, 1, 0);
  STAGE_B19(bn, 1, 1);

  for (int rep = 0; rep < tiles; rep++) {
    if (PERSIST && do_remap != 2 &&
        (int)blockIdx.x + rep * (int)gridDim.x >= nwg)
      break;
    f32x4 acc[MFR][NFR] = {};
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

#define PHASE_MFMA19(q)                                                     \
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
  } while (0)

#define READ_A19(At, q)                                                     \
  _Pragma("unroll") for (int g = 0; g < 2; g++) {                           \
    const int k16 = (g << 2) + (lane >> 4);                                 \
    _Pragma("unroll") for (int m = 0; m < 2; m++) afrag[m][g] =             \
        frag_read(At, arow_base + (q) * 32 + m * FRAG, k16);                \
  }

#define READ_B19(Bt)                                                        \
  _Pragma("unroll") for (int g = 0; g < 2; g++) {                           \
    const int k16 = (g << 2) + (lane >> 4);                                 \
    _Pragma("unroll") for (int n = 0; n < NFR; n++) bfragT[n][g] =          \
        frag_read(Bt, brow_base + n * FRAG, k16);                           \
  }

#define KTILE19(t)                                                          \
  do {                                                                      \
    const __bf16* At = A19((t) & 1);                                        \
    const __bf16* Bt = B19((t) & 1);                                        \
    READ_B19(Bt);                                                           \
    READ_A19(At, 0);                                                        \
    if ((t) + 1 < KT) STAGE_A19(bm, (t) + 1, 0);                            \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA19(0);                                                        \
    READ_A19(At, 1);                                                        \
    if ((t) + 1 < KT) STAGE_A19(bm, (t) + 1, 1);                            \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA19(1);                                                        \
    READ_A19(At, 2);                                                        \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA19(2);                                                        \
    READ_A19(At, 3);                                                        \
    /* COUNTED publish (replaces the per-tile vmcnt(0) drain): the next  */ \
    /* tile's LDS reads need A(t+1) [issued ph1-2, 4 loads/wave] and     */ \
    /* B(t+1) [issued one tile earlier, strictly older] landed.  After   */ \
    /* issuing B(t+2)'s 4 loads here, the per-wave VMEM queue newest-    */ \
    /* first is [B(t+2):4, A(t+1):4, <older, incl B(t+1)>]; vmcnt(4)     */ \
    /* waits until only B(t+2) remains in flight => everything the next  */ \
    /* tile reads has retired to LDS, while B(t+2) still spans the       */ \
    /* boundary.  r01's nondeterministic counted variant waited BEFORE   */ \
    /* the B(t+2) issues with exactly 4 outstanding — a no-op wait       */ \
    /* (profiles/r01_t256_det2.log), which this placement fixes; the     */ \
    /* 8-run multi-shape determinism screen guards the claim.            */ \
    if ((t) + 2 < KT) {                                                     \
      STAGE_B19(bn, (t) + 2, 0);                                            \
      STAGE_B19(bn, (t) + 2, 1);                                            \
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");                      \
    } else {                                                                \
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");                      \
    }                                                                       \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA19(3);                                                        \
  } while (0)

    for (long it = 0; it < KT / 2; ++it) {
      KTILE19(2 * it);
      KTILE19(2 * it + 1);
    }

    const long ebm = bm, ebn = bn;
    // advance to next tile and issue its prologue BEFORE the epilogue:
    // the LDS buffers are fully consumed (every wave drained its reads
    // via lgkmcnt(0) before its last barrier), but another wave may
    // still be inside the last phase -> barrier first.
    if (PERSIST && rep + 1 < tiles &&
        (do_remap == 2 ||
         (int)blockIdx.x + (rep + 1) * (int)gridDim.x < nwg)) {
      __builtin_amdgcn_s_barrier();
      int orig = do_remap == 2
                     ? chunk_start + rep + 1
                     : (int)blockIdx.x + (rep + 1) * (int)gridDim.x;
      if (do_remap == 1) {
        int q = nwg >> 3, r = nwg & 7;
        int xcd = orig & 7, lid = orig >> 3;
        orig = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + lid;
      }
      bm = (long)(orig / nbx) * BM;
      bn = (long)(orig % nbx) * BN;
      STAGE_A19(bm, 0, 0);
      STAGE_A19(bm, 0, 1);
      STAGE_B19(bn, 0, 0);
      STAGE_B19(bn, 0, 1);
      STAGE_B19(bn, 1, 0);
      STAGE_B19(bn, 1, 1);
    }

    const long crow_base = ebm + waveM * WM + 4 * (lane >> 4);
    const long ccol_base = ebn + waveN * WN + (lane & 15);
    const bool interior = (ebm + BM <= M) && (ebn + BN <= N);
    if (interior) {
#pragma unroll
      for (int m = 0; m < MFR; m++) {
#pragma unroll
        for (int n = 0; n < NFR; n++) {
          const long col = ccol_base + n * FRAG;
#pragma unroll
          for (int r = 0; r < 4; r++) {
            const long row = crow_base + m * FRAG + r;
            if (c_is_bf16)
              ((unsigned short*)C)[row * N + col] = bf16_rne(acc[m][n][r]);
            else
              ((float*)C)[row * N + col] = acc[m][n][r];
          }
        }
      }
    } else {
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
            if (c_is_bf16)
              ((unsigned short*)C)[row * N + col] = bf16_rne(acc[m][n][r]);
            else
              ((float*)C)[row * N + col] = acc[m][n][r];
          }
        }
      }
    }
  }
#undef A19
#undef B19
#undef STAGE_A19
#undef STAGE_B19
#undef PHASE_MFMA19
#undef READ_A19
#undef READ_B19
#undef KTILE19
}

}  // namespace

extern "C" int cc_gemm_v19(const void* A, const void* B, void* C, long M,
                           long N, long K, int c_is_bf16, int persist,
                           int remap, unsigned long long stream) {
  if (K % (2 * BK) != 0 || K / BK < 4) return -2;
  int nbx = (int)((N + BN - 1) / BN);
  int nby = (int)((M + BM - 1) / BM);
  int nwg = nbx * nby;
  int grid = nwg;
  if (persist) grid = nwg < 256 ? nwg : 256;
  if (persist)
    hipLaunchKernelGGL((k_gemm_v19<1>), dim3(grid), dim3(512), 0,
                       (hipStream_t)stream, (const __bf16*)A,
                       (const __bf16*)B, C, M, N, K, c_is_bf16, nbx, nwg,
                       remap);
  else
    hipLaunchKernelGGL((k_gemm_v19<0>), dim3(grid), dim3(512), 0,
                       (hipStream_t)stream, (const __bf16*)A,
                       (const __bf16*)B, C, M, N, K, c_is_bf16, nbx, nwg,
                       remap);
  return hipGetLastError() == hipSuccess ? 0 : -1;
}

// ---- V20: v19 counted publish + v17 one-phase-ahead afrag pipeline ----
// v13 showed register staging costs more than the glds path's vmcnt(0)
// drain (704 vs 791 at patch64), but persistence + remap wins at the
// skinny-N ViT shapes (fc1 784 vs 705).  v15 keeps the production
// KTILE2 glds schedule verbatim and adds the persistent outer loop,
// issuing the NEXT tile's 12-glds prologue BEFORE the epilogue stores
// so its HBM latency hides under the C writeback.
namespace {

template <int PERSIST>
__global__ __launch_bounds__(512, 1) void k_gemm_v20(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ C, long M, long N, long K, int c_is_bf16, int nbx,
    int nwg, int do_remap) {
  __shared__ __bf16 lds[2 * (BM + BN) * BK];  // 128 KiB
#define A20(b) (lds + (b) * (BM * BK))
#define B20(b) (lds + 2 * (BM * BK) + (b) * (BN * BK))
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int waveM = wid >> 2, waveN = wid & 3;
  const long KT = K / BK;  // caller guarantees even, >= 4
  const long srow = wid * 16;
  const int arow_base = waveM * WM + (lane & 15);
  const int brow_base = waveN * WN + (lane & 15);
  bf16x8 bfragT[NFR][2];
  bf16x8 afrag[2][2][2];    // [pipeline buf][m][kg]

#define STAGE_A20(bmv, t, h)                                                \
  stage_half15(A, K, (bmv) + (h) * 128 + srow, M, (t) * BK,                 \
               A20((t) & 1) + ((h) * 128 + srow) * BK, lane)
#define STAGE_B20(bnv, t, h)                                                \
  stage_half15(B, K, (bnv) + (h) * 128 + srow, N, (t) * BK,                 \
               B20((t) & 1) + ((h) * 128 + srow) * BK, lane)

  // tile index for rep r of this WG under the three order modes:
  //   0: orig = bid + r*grid                  (stride)
  //   1: xcd-remap(orig) of mode 0            (stride + L2 affinity)
  //   2: orig = chunk_start(bid) + r          (contiguous row-major chunk:
  //      the A row band stays hot in L2 across ~nbx consecutive reps)
  int chunk_start = 0, chunk_len = 1;
  if (PERSIST && do_remap == 2) {
    int q = nwg / (int)gridDim.x, r = nwg % (int)gridDim.x;
    int b = (int)blockIdx.x;
    chunk_len = q + (b < r ? 1 : 0);
    chunk_start = b * q + (b < r ? b : r);
  }
  const int tiles =
      PERSIST ? (do_remap == 2 ? chunk_len
                               : (nwg + (int)gridDim.x - 1) / (int)gridDim.x)
              : 1;
  long bm = 0, bn = 0;
  {
    int orig = do_remap == 2 ? chunk_start : (int)blockIdx.x;
    if (do_remap == 1) {
      int q = nwg >> 3, r = nwg & 7;
      int xcd = orig & 7, lid = orig >> 3;
      orig = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + lid;
    }
    bm = (long)(orig / nbx) * BM;
    bn = (long)(orig % nbx) * BN;
  }
  // first prologue (full-latency; later ones hide under epilogues)
  STAGE_A20(bm, 0, 0);
  STAGE_A20(bm, 0, 1);
  STAGE_B20(bn, 0, 0);
  STAGE_B20(bn, 0, 1);
  STAGE_B20(bn, 1, 0);
  STAGE_B20(bn, 1, 1);

  for (int rep = 0; rep < tiles; rep++) {
    if (PERSIST && do_remap != 2 &&
        (int)blockIdx.x + rep * (int)gridDim.x >= nwg)
      break;
    f32x4 acc[MFR][NFR] = {};
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

#define PHASE_MFMA20(q, pb)                                                 \
  do {                                                                      \
    __builtin_amdgcn_s_setprio(1);                                          \
    _Pragma("unroll") for (int g = 0; g < 2; g++) {                         \
      _Pragma("unroll") for (int m = 0; m < 2; m++) {                       \
        _Pragma("unroll") for (int n = 0; n < NFR; n++) {                   \
          acc[2 * (q) + m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(    \
              afrag[pb][m][g], bfragT[n][g], acc[2 * (q) + m][n], 0, 0, 0); \
        }                                                                   \
      }                                                                     \
    }                                                                       \
    __builtin_amdgcn_s_setprio(0);                                          \
  } while (0)

#define READ_A20(At, q, pb)                                                 \
  _Pragma("unroll") for (int g = 0; g < 2; g++) {                           \
    const int k16 = (g << 2) + (lane >> 4);                                 \
    _Pragma("unroll") for (int m = 0; m < 2; m++) afrag[pb][m][g] =         \
        frag_read(At, arow_base + (q) * 32 + m * FRAG, k16);                \
  }

#define READ_B20(Bt)                                                        \
  _Pragma("unroll") for (int g = 0; g < 2; g++) {                           \
    const int k16 = (g << 2) + (lane >> 4);                                 \
    _Pragma("unroll") for (int n = 0; n < NFR; n++) bfragT[n][g] =          \
        frag_read(Bt, brow_base + n * FRAG, k16);                           \
  }

#define KTILE20(t)                                                          \
  do {                                                                      \
    const __bf16* At = A20((t) & 1);                                        \
    const __bf16* Bt = B20((t) & 1);                                        \
    READ_B20(Bt);                                                           \
    READ_A20(At, 0, 0);                                                     \
    if ((t) + 1 < KT) STAGE_A20(bm, (t) + 1, 0);                            \
    __builtin_amdgcn_s_barrier();                                           \
    READ_A20(At, 1, 1); /* next quadrant BEFORE this cluster */             \
    PHASE_MFMA20(0, 0);                                                     \
    if ((t) + 1 < KT) STAGE_A20(bm, (t) + 1, 1);                            \
    __builtin_amdgcn_s_barrier();                                           \
    READ_A20(At, 2, 0);                                                     \
    PHASE_MFMA20(1, 1);                                                     \
    __builtin_amdgcn_s_barrier();                                           \
    READ_A20(At, 3, 1);                                                     \
    PHASE_MFMA20(2, 0);                                                     \
    if ((t) + 2 < KT) {                                                     \
      STAGE_B20(bn, (t) + 2, 0);                                            \
      STAGE_B20(bn, (t) + 2, 1);                                            \
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");                      \
    } else {                                                                \
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");                      \
    }                                                                       \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA20(3, 1);                                                     \
  } while (0)

    for (long it = 0; it < KT / 2; ++it) {
      KTILE20(2 * it);
      KTILE20(2 * it + 1);
    }

    const long ebm = bm, ebn = bn;
    // advance to next tile and issue its prologue BEFORE the epilogue:
    // the LDS buffers are fully consumed (every wave drained its reads
    // via lgkmcnt(0) before its last barrier), but another wave may
    // still be inside the last phase -> barrier first.
    if (PERSIST && rep + 1 < tiles &&
        (do_remap == 2 ||
         (int)blockIdx.x + (rep + 1) * (int)gridDim.x < nwg)) {
      __builtin_amdgcn_s_barrier();
      int orig = do_remap == 2
                     ? chunk_start + rep + 1
                     : (int)blockIdx.x + (rep + 1) * (int)gridDim.x;
      if (do_remap == 1) {
        int q = nwg >> 3, r = nwg & 7;
        int xcd = orig & 7, lid = orig >> 3;
        orig = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + lid;
      }
      bm = (long)(orig / nbx) * BM;
      bn = (long)(orig % nbx) * BN;
      STAGE_A20(bm, 0, 0);
      STAGE_A20(bm, 0, 1);
      STAGE_B20(bn, 0, 0);
      STAGE_B20(bn, 0, 1);
      STAGE_B20(bn, 1, 0);
      STAGE_B20(bn, 1, 1);
    }

    const long crow_base = ebm + waveM * WM + 4 * (lane >> 4);
    const long ccol_base = ebn + waveN * WN + (lane & 15);
    const bool interior = (ebm + BM <= M) && (ebn + BN <= N);
    if (interior) {
#pragma unroll
      for (int m = 0; m < MFR; m++) {
#pragma unroll
        for (int n = 0; n < NFR; n++) {
          const long col = ccol_base + n * FRAG;
#pragma unroll
          for (int r = 0; r < 4; r++) {
            const long row = crow_base + m * FRAG + r;
            if (c_is_bf16)
              ((unsigned short*)C)[row * N + col] = bf16_rne(acc[m][n][r]);
            else
              ((float*)C)[row * N + col] = acc[m][n][r];
          }
        }
      }
    } else {
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
            if (c_is_bf16)
              ((unsigned short*)C)[row * N + col] = bf16_rne(acc[m][n][r]);
            else
              ((float*)C)[row * N + col] = acc[m][n][r];
          }
        }
      }
    }
  }
#undef A20
#undef B20
#undef STAGE_A20
#undef STAGE_B20
#undef PHASE_MFMA20
#undef READ_A20
#undef READ_B20
#undef KTILE20
}

}  // namespace

extern "C" int cc_gemm_v20(const void* A, const void* B, void* C, long M,
                           long N, long K, int c_is_bf16, int persist,
                           int remap, unsigned long long stream) {
  if (K % (2 * BK) != 0 || K / BK < 4) return -2;
  int nbx = (int)((N + BN - 1) / BN);
  int nby = (int)((M + BM - 1) / BM);
  int nwg = nbx * nby;
  int grid = nwg;
  if (persist) grid = nwg < 256 ? nwg : 256;
  if (persist)
    hipLaunchKernelGGL((k_gemm_v20<1>), dim3(grid), dim3(512), 0,
                       (hipStream_t)stream, (const __bf16*)A,
                       (const __bf16*)B, C, M, N, K, c_is_bf16, nbx, nwg,
                       remap);
  else
    hipLaunchKernelGGL((k_gemm_v20<0>), dim3(grid), dim3(512), 0,
                       (hipStream_t)stream, (const __bf16*)A,
                       (const __bf16*)B, C, M, N, K, c_is_bf16, nbx, nwg,
                       remap);
  return hipGetLastError() == hipSuccess ? 0 : -1;
}

// ---- V21: v19 + T19 sched_group_barrier MFMA/ds_read interleave (no setprio) ----
// v13 showed register staging costs more than the glds path's vmcnt(0)
// drain (704 vs 791 at patch64), but persistence + remap wins at the
// skinny-N ViT shapes (fc1 784 vs 705).  v15 keeps the production
// KTILE2 glds schedule verbatim and adds the persistent outer loop,
// issuing the NEXT tile's 12-glds prologue BEFORE the epilogue stores
// so its HBM latency hides under the C writeback.
namespace {

template <int PERSIST>
__global__ __launch_bounds__(512, 1) void k_gemm_v21(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ C, long M, long N, long K, int c_is_bf16, int nbx,
    int nwg, int do_remap) {
  __shared__ __bf16 lds[2 * (BM + BN) * BK];  // 128 KiB
#define A21(b) (lds + (b) * (BM * BK))
#define B21(b) (lds + 2 * (BM * BK) + (b) * (BN * BK))
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int waveM = wid >> 2, waveN = wid & 3;
  const long KT = K / BK;  // caller guarantees even, >= 4
  const long srow = wid * 16;
  const int arow_base = waveM * WM + (lane & 15);
  const int brow_base = waveN * WN + (lane & 15);
  bf16x8 bfragT[NFR][2];
  bf16x8 afrag[2][2];

#define STAGE_A21(bmv, t, h)                                                \
  stage_half15(A, K, (bmv) + (h) * 128 + srow, M, (t) * BK,                 \
               A21((t) & 1) + ((h) * 128 + srow) * BK, lane)
#define STAGE_B21(bnv, t, h)                                                \
  stage_half15(B, K, (bnv) + (h) * 128 + srow, N, (t) * BK,                 \
               B21((t) & 1) + ((h) * 128 + srow) * BK, lane)

  // tile index for rep r of this WG under the three order modes:
  //   0: orig = bid + r*grid                  (stride)
  //   1: xcd-remap(orig) of mode 0            (stride + L2 affinity)
  //   2: orig = chunk_start(bid) + r          (contiguous row-major chunk:
  //      the A row band stays hot in L2 across ~nbx consecutive reps)
  int chunk_start = 0, chunk_len = 1;
  if (PERSIST && do_remap == 2) {
    int q = nwg / (int)gridDim.x, r = nwg % (int)gridDim.x;
    int b = (int)blockIdx.x;
    chunk_len = q + (b < r ? 1 : 0);
    chunk_start = b * q + (b < r ? b : r);
  }
  const int tiles =
      PERSIST ? (do_remap == 2 ? chunk_len
                               : (nwg + (int)gridDim.x - 1) / (int)gridDim.x)
              : 1;
  long bm = 0, bn = 0;
  {
    int orig = do_remap == 2 ? chunk_start : (int)blockIdx.x;
    if (do_remap == 1) {
      int q = nwg >> 3, r = nwg & 7;
      int xcd = orig & 7, lid = orig >> 3;
      orig = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + lid;
    }
    bm = (long)(orig / nbx) * BM;
    bn = (long)(orig % nbx) * BN;
  }
  // first prologue (full-latency; later ones hide under epilogues)
  STAGE_A21(bm, 0, 0);
  STAGE_A21(bm, 0, 1);
  STAGE_B21(bn, 0, 0);
  STAGE_B21(bn, 0, 1);
  STAGE_B21(bn, 1, 0);
  STAGE_B21(bn, 1, 1);

  for (int rep = 0; rep < tiles; rep++) {
    if (PERSIST && do_remap != 2 &&
        (int)blockIdx.x + rep * (int)gridDim.x >= nwg)
      break;
    f32x4 acc[MFR][NFR] = {};
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

#define PHASE_MFMA21(q)                                                     \
  do {                                                                      \
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");                      \
    /* T19 dictation for this barrier region: weave the NEXT phase's 4  */ \
    /* ds_reads (and the glds issues) between the 16 MFMAs instead of   */ \
    /* after them, so their latency hides under the cluster.            */ \
    _Pragma("unroll") for (int gi = 0; gi < 4; gi++) {                      \
      __builtin_amdgcn_sched_group_barrier(0x8 /*MFMA*/, 4, 0);             \
      __builtin_amdgcn_sched_group_barrier(0x100 /*DS_READ*/, 1, 0);        \
      __builtin_amdgcn_sched_group_barrier(0x10 /*VMEM*/, 1, 0);            \
    }                                                                       \
    _Pragma("unroll") for (int g = 0; g < 2; g++) {                         \
      _Pragma("unroll") for (int m = 0; m < 2; m++) {                       \
        _Pragma("unroll") for (int n = 0; n < NFR; n++) {                   \
          acc[2 * (q) + m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(    \
              afrag[m][g], bfragT[n][g], acc[2 * (q) + m][n], 0, 0, 0);     \
        }                                                                   \
      }                                                                     \
    }                                                                       \
  } while (0)

#define READ_A21(At, q)                                                     \
  _Pragma("unroll") for (int g = 0; g < 2; g++) {                           \
    const int k16 = (g << 2) + (lane >> 4);                                 \
    _Pragma("unroll") for (int m = 0; m < 2; m++) afrag[m][g] =             \
        frag_read(At, arow_base + (q) * 32 + m * FRAG, k16);                \
  }

#define READ_B21(Bt)                                                        \
  _Pragma("unroll") for (int g = 0; g < 2; g++) {                           \
    const int k16 = (g << 2) + (lane >> 4);                                 \
    _Pragma("unroll") for (int n = 0; n < NFR; n++) bfragT[n][g] =          \
        frag_read(Bt, brow_base + n * FRAG, k16);                           \
  }

#define KTILE21(t)                                                          \
  do {                                                                      \
    const __bf16* At = A21((t) & 1);                                        \
    const __bf16* Bt = B21((t) & 1);                                        \
    READ_B21(Bt);                                                           \
    READ_A21(At, 0);                                                        \
    if ((t) + 1 < KT) STAGE_A21(bm, (t) + 1, 0);                            \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA21(0);                                                        \
    READ_A21(At, 1);                                                        \
    if ((t) + 1 < KT) STAGE_A21(bm, (t) + 1, 1);                            \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA21(1);                                                        \
    READ_A21(At, 2);                                                        \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA21(2);                                                        \
    READ_A21(At, 3);                                                        \
    /* COUNTED publish (replaces the per-tile vmcnt(0) drain): the next  */ \
    /* tile's LDS reads need A(t+1) [issued ph1-2, 4 loads/wave] and     */ \
    /* B(t+1) [issued one tile earlier, strictly older] landed.  After   */ \
    /* issuing B(t+2)'s 4 loads here, the per-wave VMEM queue newest-    */ \
    /* first is [B(t+2):4, A(t+1):4, <older, incl B(t+1)>]; vmcnt(4)     */ \
    /* waits until only B(t+2) remains in flight => everything the next  */ \
    /* tile reads has retired to LDS, while B(t+2) still spans the       */ \
    /* boundary.  r01's nondeterministic counted variant waited BEFORE   */ \
    /* the B(t+2) issues with exactly 4 outstanding — a no-op wait       */ \
    /* (profiles/r01_t256_det2.log), which this placement fixes; the     */ \
    /* 8-run multi-shape determinism screen guards the claim.            */ \
    if ((t) + 2 < KT) {                                                     \
      STAGE_B21(bn, (t) + 2, 0);                                            \
      STAGE_B21(bn, (t) + 2, 1);                                            \
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");                      \
    } else {                                                                \
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");                      \
    }                                                                       \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA21(3);                                                        \
  } while (0)

    for (long it = 0; it < KT / 2; ++it) {
      KTILE21(2 * it);
      KTILE21(2 * it + 1);
    }

    const long ebm = bm, ebn = bn;
    // advance to next tile and issue its prologue BEFORE the epilogue:
    // the LDS buffers are fully consumed (every wave drained its reads
    // via lgkmcnt(0) before its last barrier), but another wave may
    // still be inside the last phase -> barrier first.
    if (PERSIST && rep + 1 < tiles &&
        (do_remap == 2 ||
         (int)blockIdx.x + (rep + 1) * (int)gridDim.x < nwg)) {
      __builtin_amdgcn_s_barrier();
      int orig = do_remap == 2
                     ? chunk_start + rep + 1
                     : (int)blockIdx.x + (rep + 1) * (int)gridDim.x;
      if (do_remap == 1) {
        int q = nwg >> 3, r = nwg & 7;
        int xcd = orig & 7, lid = orig >> 3;
        orig = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + lid;
      }
      bm = (long)(orig / nbx) * BM;
      bn = (long)(orig % nbx) * BN;
      STAGE_A21(bm, 0, 0);
      STAGE_A21(bm, 0, 1);
      STAGE_B21(bn, 0, 0);
      STAGE_B21(bn, 0, 1);
      STAGE_B21(bn, 1, 0);
      STAGE_B21(bn, 1, 1);
    }

    const long crow_base = ebm + waveM * WM + 4 * (lane >> 4);
    const long ccol_base = ebn + waveN * WN + (lane & 15);
    const bool interior = (ebm + BM <= M) && (ebn + BN <= N);
    if (interior) {
#pragma unroll
      for (int m = 0; m < MFR; m++) {
#pragma unroll
        for (int n = 0; n < NFR; n++) {
          const long col = ccol_base + n * FRAG;
#pragma unroll
          for (int r = 0; r < 4; r++) {
            const long row = crow_base + m * FRAG + r;
            if (c_is_bf16)
              ((unsigned short*)C)[row * N + col] = bf16_rne(acc[m][n][r]);
            else
              ((float*)C)[row * N + col] = acc[m][n][r];
          }
        }
      }
    } else {
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
            if (c_is_bf16)
              ((unsigned short*)C)[row * N + col] = bf16_rne(acc[m][n][r]);
            else
              ((float*)C)[row * N + col] = acc[m][n][r];
          }
        }
      }
    }
  }
#undef A21
#undef B21
#undef STAGE_A21
#undef STAGE_B21
#undef PHASE_MFMA21
#undef READ_A21
#undef READ_B21
#undef KTILE21
}

}  // namespace

extern "C" int cc_gemm_v21(const void* A, const void* B, void* C, long M,
                           long N, long K, int c_is_bf16, int persist,
                           int remap, unsigned long long stream) {
  if (K % (2 * BK) != 0 || K / BK < 4) return -2;
  int nbx = (int)((N + BN - 1) / BN);
  int nby = (int)((M + BM - 1) / BM);
  int nwg = nbx * nby;
  int grid = nwg;
  if (persist) grid = nwg < 256 ? nwg : 256;
  if (persist)
    hipLaunchKernelGGL((k_gemm_v21<1>), dim3(grid), dim3(512), 0,
                       (hipStream_t)stream, (const __bf16*)A,
                       (const __bf16*)B, C, M, N, K, c_is_bf16, nbx, nwg,
                       remap);
  else
    hipLaunchKernelGGL((k_gemm_v21<0>), dim3(grid), dim3(512), 0,
                       (hipStream_t)stream, (const __bf16*)A,
                       (const __bf16*)B, C, M, N, K, c_is_bf16, nbx, nwg,
                       remap);
  return hipGetLastError() == hipSuccess ? 0 : -1;
}

// ---- V22: T19 pattern sweep (template<int PAT>) ----
// v13 showed register staging costs more than the glds path's vmcnt(0)
// drain (704 vs 791 at patch64), but persistence + remap wins at the
// skinny-N ViT shapes (fc1 784 vs 705).  v15 keeps the production
// KTILE2 glds schedule verbatim and adds the persistent outer loop,
// issuing the NEXT tile's 12-glds prologue BEFORE the epilogue stores
// so its HBM latency hides under the C writeback.
namespace {

template <int PERSIST, int PAT>
__global__ __launch_bounds__(512, 1) void k_gemm_v22(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ C, long M, long N, long K, int c_is_bf16, int nbx,
    int nwg, int do_remap) {
  __shared__ __bf16 lds[2 * (BM + BN) * BK];  // 128 KiB
#define A22(b) (lds + (b) * (BM * BK))
#define B22(b) (lds + 2 * (BM * BK) + (b) * (BN * BK))
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int waveM = wid >> 2, waveN = wid & 3;
  const long KT = K / BK;  // caller guarantees even, >= 4
  const long srow = wid * 16;
  const int arow_base = waveM * WM + (lane & 15);
  const int brow_base = waveN * WN + (lane & 15);
  bf16x8 bfragT[NFR][2];
  bf16x8 afrag[2][2];

#define STAGE_A22(bmv, t, h)                                                \
  stage_half15(A, K, (bmv) + (h) * 128 + srow, M, (t) * BK,                 \
               A22((t) & 1) + ((h) * 128 + srow) * BK, lane)
#define STAGE_B22(bnv, t, h)                                                \
  stage_half15(B, K, (bnv) + (h) * 128 + srow, N, (t) * BK,                 \
               B22((t) & 1) + ((h) * 128 + srow) * BK, lane)

  // tile index for rep r of this WG under the three order modes:
  //   0: orig = bid + r*grid                  (stride)
  //   1: xcd-remap(orig) of mode 0            (stride + L2 affinity)
  //   2: orig = chunk_start(bid) + r          (contiguous row-major chunk:
  //      the A row band stays hot in L2 across ~nbx consecutive reps)
  int chunk_start = 0, chunk_len = 1;
  if (PERSIST && do_remap == 2) {
    int q = nwg / (int)gridDim.x, r = nwg % (int)gridDim.x;
    int b = (int)blockIdx.x;
    chunk_len = q + (b < r ? 1 : 0);
    chunk_start = b * q + (b < r ? b : r);
  }
  const int tiles =
      PERSIST ? (do_remap == 2 ? chunk_len
                               : (nwg + (int)gridDim.x - 1) / (int)gridDim.x)
              : 1;
  long bm = 0, bn = 0;
  {
    int orig = do_remap == 2 ? chunk_start : (int)blockIdx.x;
    if (do_remap == 1) {
      int q = nwg >> 3, r = nwg & 7;
      int xcd = orig & 7, lid = orig >> 3;
      orig = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + lid;
    }
    bm = (long)(orig / nbx) * BM;
    bn = (long)(orig % nbx) * BN;
  }
  // first prologue (full-latency; later ones hide under epilogues)
  STAGE_A22(bm, 0, 0);
  STAGE_A22(bm, 0, 1);
  STAGE_B22(bn, 0, 0);
  STAGE_B22(bn, 0, 1);
  STAGE_B22(bn, 1, 0);
  STAGE_B22(bn, 1, 1);

  for (int rep = 0; rep < tiles; rep++) {
    if (PERSIST && do_remap != 2 &&
        (int)blockIdx.x + rep * (int)gridDim.x >= nwg)
      break;
    f32x4 acc[MFR][NFR] = {};
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

#define PHASE_MFMA22(q)                                                     \
  do {                                                                      \
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");                      \
    /* T19 dictation for this barrier region: weave the NEXT phase's 4  */ \
    /* ds_reads (and the glds issues) between the 16 MFMAs instead of   */ \
    /* after them, so their latency hides under the cluster.            */ \
    if constexpr (PAT == 0) {                                               \
      _Pragma("unroll") for (int gi = 0; gi < 4; gi++) {                    \
        __builtin_amdgcn_sched_group_barrier(0x8, 4, 0);                    \
        __builtin_amdgcn_sched_group_barrier(0x100, 1, 0);                  \
        __builtin_amdgcn_sched_group_barrier(0x10, 1, 0);                   \
      }                                                                     \
    } else if constexpr (PAT == 1) {                                        \
      _Pragma("unroll") for (int gi = 0; gi < 8; gi++) {                    \
        __builtin_amdgcn_sched_group_barrier(0x8, 2, 0);                    \
        __builtin_amdgcn_sched_group_barrier(0x100, 1, 0);                  \
        __builtin_amdgcn_sched_group_barrier(0x10, 1, 0);                   \
      }                                                                     \
    } else if constexpr (PAT == 2) {                                        \
      _Pragma("unroll") for (int gi = 0; gi < 4; gi++) {                    \
        __builtin_amdgcn_sched_group_barrier(0x8, 4, 0);                    \
        __builtin_amdgcn_sched_group_barrier(0x100, 2, 0);                  \
      }                                                                     \
      __builtin_amdgcn_sched_group_barrier(0x10, 4, 0);                     \
    } else if constexpr (PAT == 3) {                                        \
      __builtin_amdgcn_sched_group_barrier(0x8, 2, 0);                      \
      _Pragma("unroll") for (int gi = 0; gi < 6; gi++) {                    \
        __builtin_amdgcn_sched_group_barrier(0x100, 1, 0);                  \
        __builtin_amdgcn_sched_group_barrier(0x8, 2, 0);                    \
        __builtin_amdgcn_sched_group_barrier(0x10, 1, 0);                   \
      }                                                                     \
    }                                                                       \
    _Pragma("unroll") for (int g = 0; g < 2; g++) {                         \
      _Pragma("unroll") for (int m = 0; m < 2; m++) {                       \
        _Pragma("unroll") for (int n = 0; n < NFR; n++) {                   \
          acc[2 * (q) + m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(    \
              afrag[m][g], bfragT[n][g], acc[2 * (q) + m][n], 0, 0, 0);     \
        }                                                                   \
      }                                                                     \
    }                                                                       \
  } while (0)

#define READ_A22(At, q)                                                     \
  _Pragma("unroll") for (int g = 0; g < 2; g++) {                           \
    const int k16 = (g << 2) + (lane >> 4);                                 \
    _Pragma("unroll") for (int m = 0; m < 2; m++) afrag[m][g] =             \
        frag_read(At, arow_base + (q) * 32 + m * FRAG, k16);                \
  }

#define READ_B22(Bt)                                                        \
  _Pragma("unroll") for (int g = 0; g < 2; g++) {                           \
    const int k16 = (g << 2) + (lane >> 4);                                 \
    _Pragma("unroll") for (int n = 0; n < NFR; n++) bfragT[n][g] =          \
        frag_read(Bt, brow_base + n * FRAG, k16);                           \
  }

#define KTILE22(t)                                                          \
  do {                                                                      \
    const __bf16* At = A22((t) & 1);                                        \
    const __bf16* Bt = B22((t) & 1);                                        \
    READ_B22(Bt);                                                           \
    READ_A22(At, 0);                                                        \
    if ((t) + 1 < KT) STAGE_A22(bm, (t) + 1, 0);                            \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA22(0);                                                        \
    READ_A22(At, 1);                                                        \
    if ((t) + 1 < KT) STAGE_A22(bm, (t) + 1, 1);                            \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA22(1);                                                        \
    READ_A22(At, 2);                                                        \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA22(2);                                                        \
    READ_A22(At, 3);                                                        \
    /* COUNTED publish (replaces the per-tile vmcnt(0) drain): the next  */ \
    /* tile's LDS reads need A(t+1) [issued ph1-2, 4 loads/wave] and     */ \
    /* B(t+1) [issued one tile earlier, strictly older] landed.  After   */ \
    /* issuing B(t+2)'s 4 loads here, the per-wave VMEM queue newest-    */ \
    /* first is [B(t+2):4, A(t+1):4, <older, incl B(t+1)>]; vmcnt(4)     */ \
    /* waits until only B(t+2) remains in flight => everything the next  */ \
    /* tile reads has retired to LDS, while B(t+2) still spans the       */ \
    /* boundary.  r01's nondeterministic counted variant waited BEFORE   */ \
    /* the B(t+2) issues with exactly 4 outstanding — a no-op wait       */ \
    /* (profiles/r01_t256_det2.log), which this placement fixes; the     */ \
    /* 8-run multi-shape determinism screen guards the claim.            */ \
    if ((t) + 2 < KT) {                                                     \
      STAGE_B22(bn, (t) + 2, 0);                                            \
      STAGE_B22(bn, (t) + 2, 1);                                            \
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");                      \
    } else {                                                                \
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");                      \
    }                                                                       \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA22(3);                                                        \
  } while (0)

    for (long it = 0; it < KT / 2; ++it) {
      KTILE22(2 * it);
      KTILE22(2 * it + 1);
    }

    const long ebm = bm, ebn = bn;
    // advance to next tile and issue its prologue BEFORE the epilogue:
    // the LDS buffers are fully consumed (every wave drained its reads
    // via lgkmcnt(0) before its last barrier), but another wave may
    // still be inside the last phase -> barrier first.
    if (PERSIST && rep + 1 < tiles &&
        (do_remap == 2 ||
         (int)blockIdx.x + (rep + 1) * (int)gridDim.x < nwg)) {
      __builtin_amdgcn_s_barrier();
      int orig = do_remap == 2
                     ? chunk_start + rep + 1
                     : (int)blockIdx.x + (rep + 1) * (int)gridDim.x;
      if (do_remap == 1) {
        int q = nwg >> 3, r = nwg & 7;
        int xcd = orig & 7, lid = orig >> 3;
        orig = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + lid;
      }
      bm = (long)(orig / nbx) * BM;
      bn = (long)(orig % nbx) * BN;
      STAGE_A22(bm, 0, 0);
      STAGE_A22(bm, 0, 1);
      STAGE_B22(bn, 0, 0);
      STAGE_B22(bn, 0, 1);
      STAGE_B22(bn, 1, 0);
      STAGE_B22(bn, 1, 1);
    }

    const long crow_base = ebm + waveM * WM + 4 * (lane >> 4);
    const long ccol_base = ebn + waveN * WN + (lane & 15);
    const bool interior = (ebm + BM <= M) && (ebn + BN <= N);
    if (interior) {
#pragma unroll
      for (int m = 0; m < MFR; m++) {
#pragma unroll
        for (int n = 0; n < NFR; n++) {
          const long col = ccol_base + n * FRAG;
#pragma unroll
          for (int r = 0; r < 4; r++) {
            const long row = crow_base + m * FRAG + r;
            if (c_is_bf16)
              ((unsigned short*)C)[row * N + col] = bf16_rne(acc[m][n][r]);
            else
              ((float*)C)[row * N + col] = acc[m][n][r];
          }
        }
      }
    } else {
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
            if (c_is_bf16)
              ((unsigned short*)C)[row * N + col] = bf16_rne(acc[m][n][r]);
            else
              ((float*)C)[row * N + col] = acc[m][n][r];
          }
        }
      }
    }
  }
#undef A22
#undef B22
#undef STAGE_A22
#undef STAGE_B22
#undef PHASE_MFMA22
#undef READ_A22
#undef READ_B22
#undef KTILE22
}

}  // namespace

extern "C" int cc_gemm_v22(const void* A, const void* B, void* C, long M,
                           long N, long K, int c_is_bf16, int pat,
                           int remap, unsigned long long stream) {
  if (K % (2 * BK) != 0 || K / BK < 4) return -2;
  int nbx = (int)((N + BN - 1) / BN);
  int nby = (int)((M + BM - 1) / BM);
  int nwg = nbx * nby;
  int grid = nwg < 256 ? nwg : 256;
#define L22(P)                                                             \
  hipLaunchKernelGGL((k_gemm_v22<1, P>), dim3(grid), dim3(512), 0,         \
                     (hipStream_t)stream, (const __bf16*)A,                \
                     (const __bf16*)B, C, M, N, K, c_is_bf16, nbx, nwg,    \
                     remap)
  switch (pat) {
    case 0: L22(0); break;
    case 1: L22(1); break;
    case 2: L22(2); break;
    case 3: L22(3); break;
    default: return -3;
  }
#undef L22
  return hipGetLastError() == hipSuccess ? 0 : -1;
}
// ---- V16: v15 + per-XCD atomic tile queues ----
// v15's static chunking leaves a straggler tail when nwg/grid has a big
// fractional part (out64: 789/256 = 3.08 -> 21 WGs run a 4th tile while
// 235 CUs idle).  v16 pops tile indices from one atomic counter per XCD
// over a contiguous row-major range: dynamic balance inside each XCD's
// 32 CUs + the contiguous window keeps A row bands hot in that XCD's L2.
// Results stay bitwise deterministic (each tile's C block is a pure
// function of A/B; only WHO computes it varies run to run).
namespace {

template <int MODE>  // 0: global queue; 1: per-XCD ranges
__global__ __launch_bounds__(512, 1) void k_gemm_v16(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ C, long M, long N, long K, int c_is_bf16, int nbx,
    int nwg, int* __restrict__ ctrs) {
  __shared__ __bf16 lds[2 * (BM + BN) * BK];  // 128 KiB
  __shared__ int next_tile_s;
#define A16(b) (lds + (b) * (BM * BK))
#define B16(b) (lds + 2 * (BM * BK) + (b) * (BN * BK))
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int waveM = wid >> 2, waveN = wid & 3;
  const long KT = K / BK;
  const long srow = wid * 16;
  const int arow_base = waveM * WM + (lane & 15);
  const int brow_base = waveN * WN + (lane & 15);
  bf16x8 bfragT[NFR][2];
  bf16x8 afrag[2][2];

  // this WG's queue: XCD id from the dispatch round-robin (T1)
  const int xcd = MODE ? ((int)blockIdx.x & 7) : 0;
  int rstart = 0, rlen = nwg;
  if (MODE) {
    const int q = nwg >> 3, r = nwg & 7;
    rlen = q + (xcd < r ? 1 : 0);
    rstart = xcd * q + (xcd < r ? xcd : r);
  }

#define STAGE_A16(bmv, t, h)                                                \
  stage_half15(A, K, (bmv) + (h) * 128 + srow, M, (t) * BK,                 \
               A16((t) & 1) + ((h) * 128 + srow) * BK, lane)
#define STAGE_B16(bnv, t, h)                                                \
  stage_half15(B, K, (bnv) + (h) * 128 + srow, N, (t) * BK,                 \
               B16((t) & 1) + ((h) * 128 + srow) * BK, lane)

#define POP_TILE(dst)                                                       \
  do {                                                                      \
    if (tid == 0) next_tile_s = atomicAdd(ctrs + xcd, 1);                   \
    __syncthreads();                                                        \
    dst = next_tile_s;                                                      \
  } while (0)

  int idx;
  POP_TILE(idx);
  if (idx >= rlen) return;
  long bm = (long)((rstart + idx) / nbx) * BM;
  long bn = (long)((rstart + idx) % nbx) * BN;
  STAGE_A16(bm, 0, 0);
  STAGE_A16(bm, 0, 1);
  STAGE_B16(bn, 0, 0);
  STAGE_B16(bn, 0, 1);
  STAGE_B16(bn, 1, 0);
  STAGE_B16(bn, 1, 1);

  while (true) {
    f32x4 acc[MFR][NFR] = {};
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

#define PHASE_MFMA16(q)                                                     \
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
  } while (0)

#define READ_A16(At, q)                                                     \
  _Pragma("unroll") for (int g = 0; g < 2; g++) {                           \
    const int k16 = (g << 2) + (lane >> 4);                                 \
    _Pragma("unroll") for (int m = 0; m < 2; m++) afrag[m][g] =             \
        frag_read(At, arow_base + (q) * 32 + m * FRAG, k16);                \
  }

#define READ_B16(Bt)                                                        \
  _Pragma("unroll") for (int g = 0; g < 2; g++) {                           \
    const int k16 = (g << 2) + (lane >> 4);                                 \
    _Pragma("unroll") for (int n = 0; n < NFR; n++) bfragT[n][g] =          \
        frag_read(Bt, brow_base + n * FRAG, k16);                           \
  }

#define KTILE16(t)                                                          \
  do {                                                                      \
    const __bf16* At = A16((t) & 1);                                        \
    const __bf16* Bt = B16((t) & 1);                                        \
    READ_B16(Bt);                                                           \
    READ_A16(At, 0);                                                        \
    if ((t) + 1 < KT) STAGE_A16(bm, (t) + 1, 0);                            \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA16(0);                                                        \
    READ_A16(At, 1);                                                        \
    if ((t) + 1 < KT) STAGE_A16(bm, (t) + 1, 1);                            \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA16(1);                                                        \
    READ_A16(At, 2);                                                        \
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");                        \
    if ((t) + 2 < KT) STAGE_B16(bn, (t) + 2, 0);                            \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA16(2);                                                        \
    READ_A16(At, 3);                                                        \
    if ((t) + 2 < KT) STAGE_B16(bn, (t) + 2, 1);                            \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA16(3);                                                        \
  } while (0)

    for (long it = 0; it < KT / 2; ++it) {
      KTILE16(2 * it);
      KTILE16(2 * it + 1);
    }

    const long ebm = bm, ebn = bn;
    // pop + stage the next tile before the epilogue (latency hides
    // under the C stores).  POP_TILE carries its own __syncthreads,
    // which also fences the LDS reuse.
    POP_TILE(idx);
    const bool more = idx < rlen;
    if (more) {
      bm = (long)((rstart + idx) / nbx) * BM;
      bn = (long)((rstart + idx) % nbx) * BN;
      STAGE_A16(bm, 0, 0);
      STAGE_A16(bm, 0, 1);
      STAGE_B16(bn, 0, 0);
      STAGE_B16(bn, 0, 1);
      STAGE_B16(bn, 1, 0);
      STAGE_B16(bn, 1, 1);
    }

    const long crow_base = ebm + waveM * WM + 4 * (lane >> 4);
    const long ccol_base = ebn + waveN * WN + (lane & 15);
    const bool interior = (ebm + BM <= M) && (ebn + BN <= N);
    if (interior) {
#pragma unroll
      for (int m = 0; m < MFR; m++) {
#pragma unroll
        for (int n = 0; n < NFR; n++) {
          const long col = ccol_base + n * FRAG;
#pragma unroll
          for (int r = 0; r < 4; r++) {
            const long row = crow_base + m * FRAG + r;
            if (c_is_bf16)
              ((unsigned short*)C)[row * N + col] = bf16_rne(acc[m][n][r]);
            else
              ((float*)C)[row * N + col] = acc[m][n][r];
          }
        }
      }
    } else {
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
            if (c_is_bf16)
              ((unsigned short*)C)[row * N + col] = bf16_rne(acc[m][n][r]);
            else
              ((float*)C)[row * N + col] = acc[m][n][r];
          }
        }
      }
    }
    if (!more) return;
  }
#undef A16
#undef B16
#undef STAGE_A16
#undef STAGE_B16
#undef PHASE_MFMA16
#undef READ_A16
#undef READ_B16
#undef KTILE16
#undef POP_TILE
}

}  // namespace

extern "C" int cc_gemm_v16(const void* A, const void* B, void* C, long M,
                           long N, long K, int c_is_bf16, int mode,
                           unsigned long long stream) {
  if (K % (2 * BK) != 0 || K / BK < 4) return -2;
  static int* ctrs = nullptr;
  if (!ctrs && hipMalloc(&ctrs, 8 * sizeof(int)) != hipSuccess) return -3;
  int nbx = (int)((N + BN - 1) / BN);
  int nby = (int)((M + BM - 1) / BM);
  int nwg = nbx * nby;
  int grid = nwg < 256 ? nwg : 256;
  hipMemsetAsync(ctrs, 0, 8 * sizeof(int), (hipStream_t)stream);
  if (mode)
    hipLaunchKernelGGL((k_gemm_v16<1>), dim3(grid), dim3(512), 0,
                       (hipStream_t)stream, (const __bf16*)A,
                       (const __bf16*)B, C, M, N, K, c_is_bf16, nbx, nwg,
                       ctrs);
  else
    hipLaunchKernelGGL((k_gemm_v16<0>), dim3(grid), dim3(512), 0,
                       (hipStream_t)stream, (const __bf16*)A,
                       (const __bf16*)B, C, M, N, K, c_is_bf16, nbx, nwg,
                       ctrs);
  return hipGetLastError() == hipSuccess ? 0 : -1;
}

// ---- V17: v15 + one-phase-ahead afrag pipeline (no lgkm drains) ----
// PMC evidence (profiles/r02 pmc_*): 38-42% of wave cycles park at
// waitcnt/barrier even in the bare kernel.  Each phase issues its
// ds_reads just before its barrier, so the post-barrier lgkmcnt(0)
// exposes most of the LDS latency.  v17 double-buffers afrag and issues
// quadrant q+1's reads BEFORE quadrant q's MFMA cluster (outside the
// setprio window — v12's mistake was inside it), dropping the explicit
// lgkm drain entirely: hipcc's register scoreboard emits counted waits
// for exactly the fragments each MFMA consumes.
// WAR safety without the lgkm0-before-barrier argument: bfragT reads
// complete before their ph-q0 MFMA consumes them (compiler wait), 2+
// phases before STAGE_B2(t+2) overwrites that buffer; A-buf writes
// target the non-read buffer.  vmcnt(0) publish moved to ph2 (after
// MFMA(1), before the B(t+2) issues).
namespace {

template <int PERSIST>
__global__ __launch_bounds__(512, 1) void k_gemm_v17(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ C, long M, long N, long K, int c_is_bf16, int nbx,
    int nwg, int do_remap) {
  __shared__ __bf16 lds[2 * (BM + BN) * BK];  // 128 KiB
#define A17(b) (lds + (b) * (BM * BK))
#define B17(b) (lds + 2 * (BM * BK) + (b) * (BN * BK))
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int waveM = wid >> 2, waveN = wid & 3;
  const long KT = K / BK;
  const long srow = wid * 16;
  const int arow_base = waveM * WM + (lane & 15);
  const int brow_base = waveN * WN + (lane & 15);
  bf16x8 bfragT[NFR][2];
  bf16x8 afrag[2][2][2];  // [pipeline buf][m][kg]

#define STAGE_A17(bmv, t, h)                                                \
  stage_half15(A, K, (bmv) + (h) * 128 + srow, M, (t) * BK,                 \
               A17((t) & 1) + ((h) * 128 + srow) * BK, lane)
#define STAGE_B17(bnv, t, h)                                                \
  stage_half15(B, K, (bnv) + (h) * 128 + srow, N, (t) * BK,                 \
               B17((t) & 1) + ((h) * 128 + srow) * BK, lane)

  const int tiles = PERSIST ? (nwg + (int)gridDim.x - 1) / (int)gridDim.x : 1;
  long bm = 0, bn = 0;
  {
    int orig = (int)blockIdx.x;
    if (do_remap) {
      int q = nwg >> 3, r = nwg & 7;
      int xcd = orig & 7, lid = orig >> 3;
      orig = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + lid;
    }
    bm = (long)(orig / nbx) * BM;
    bn = (long)(orig % nbx) * BN;
  }
  STAGE_A17(bm, 0, 0);
  STAGE_A17(bm, 0, 1);
  STAGE_B17(bn, 0, 0);
  STAGE_B17(bn, 0, 1);
  STAGE_B17(bn, 1, 0);
  STAGE_B17(bn, 1, 1);

  for (int rep = 0; rep < tiles; rep++) {
    if (PERSIST && (int)blockIdx.x + rep * (int)gridDim.x >= nwg) break;
    f32x4 acc[MFR][NFR] = {};
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

#define MFMA17(q, pb)                                                       \
  do {                                                                      \
    __builtin_amdgcn_s_setprio(1);                                          \
    _Pragma("unroll") for (int g = 0; g < 2; g++) {                         \
      _Pragma("unroll") for (int m = 0; m < 2; m++) {                       \
        _Pragma("unroll") for (int n = 0; n < NFR; n++) {                   \
          acc[2 * (q) + m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(    \
              afrag[pb][m][g], bfragT[n][g], acc[2 * (q) + m][n], 0, 0, 0); \
        }                                                                   \
      }                                                                     \
    }                                                                       \
    __builtin_amdgcn_s_setprio(0);                                          \
  } while (0)

#define READ_A17(At, q, pb)                                                 \
  _Pragma("unroll") for (int g = 0; g < 2; g++) {                           \
    const int k16 = (g << 2) + (lane >> 4);                                 \
    _Pragma("unroll") for (int m = 0; m < 2; m++) afrag[pb][m][g] =         \
        frag_read(At, arow_base + (q) * 32 + m * FRAG, k16);                \
  }

#define READ_B17(Bt)                                                        \
  _Pragma("unroll") for (int g = 0; g < 2; g++) {                           \
    const int k16 = (g << 2) + (lane >> 4);                                 \
    _Pragma("unroll") for (int n = 0; n < NFR; n++) bfragT[n][g] =          \
        frag_read(Bt, brow_base + n * FRAG, k16);                           \
  }

#define KTILE17(t)                                                          \
  do {                                                                      \
    const __bf16* At = A17((t) & 1);                                        \
    const __bf16* Bt = B17((t) & 1);                                        \
    READ_B17(Bt);                                                           \
    READ_A17(At, 0, 0);                                                     \
    if ((t) + 1 < KT) STAGE_A17(bm, (t) + 1, 0);                            \
    __builtin_amdgcn_s_barrier();                                           \
    READ_A17(At, 1, 1); /* next quadrant BEFORE this cluster */             \
    MFMA17(0, 0);                                                           \
    if ((t) + 1 < KT) STAGE_A17(bm, (t) + 1, 1);                            \
    __builtin_amdgcn_s_barrier();                                           \
    READ_A17(At, 2, 0);                                                     \
    MFMA17(1, 1);                                                           \
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");                        \
    if ((t) + 2 < KT) STAGE_B17(bn, (t) + 2, 0);                            \
    __builtin_amdgcn_s_barrier();                                           \
    READ_A17(At, 3, 1);                                                     \
    MFMA17(2, 0);                                                           \
    if ((t) + 2 < KT) STAGE_B17(bn, (t) + 2, 1);                            \
    __builtin_amdgcn_s_barrier();                                           \
    MFMA17(3, 1);                                                           \
  } while (0)

    for (long it = 0; it < KT / 2; ++it) {
      KTILE17(2 * it);
      KTILE17(2 * it + 1);
    }

    const long ebm = bm, ebn = bn;
    if (PERSIST && rep + 1 < tiles &&
        (int)blockIdx.x + (rep + 1) * (int)gridDim.x < nwg) {
      __builtin_amdgcn_s_barrier();
      int orig = (int)blockIdx.x + (rep + 1) * (int)gridDim.x;
      if (do_remap) {
        int q = nwg >> 3, r = nwg & 7;
        int xcd = orig & 7, lid = orig >> 3;
        orig = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + lid;
      }
      bm = (long)(orig / nbx) * BM;
      bn = (long)(orig % nbx) * BN;
      STAGE_A17(bm, 0, 0);
      STAGE_A17(bm, 0, 1);
      STAGE_B17(bn, 0, 0);
      STAGE_B17(bn, 0, 1);
      STAGE_B17(bn, 1, 0);
      STAGE_B17(bn, 1, 1);
    }

    const long crow_base = ebm + waveM * WM + 4 * (lane >> 4);
    const long ccol_base = ebn + waveN * WN + (lane & 15);
    const bool interior = (ebm + BM <= M) && (ebn + BN <= N);
    if (interior) {
#pragma unroll
      for (int m = 0; m < MFR; m++) {
#pragma unroll
        for (int n = 0; n < NFR; n++) {
          const long col = ccol_base + n * FRAG;
#pragma unroll
          for (int r = 0; r < 4; r++) {
            const long row = crow_base + m * FRAG + r;
            if (c_is_bf16)
              ((unsigned short*)C)[row * N + col] = bf16_rne(acc[m][n][r]);
            else
              ((float*)C)[row * N + col] = acc[m][n][r];
          }
        }
      }
    } else {
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
            if (c_is_bf16)
              ((unsigned short*)C)[row * N + col] = bf16_rne(acc[m][n][r]);
            else
              ((float*)C)[row * N + col] = acc[m][n][r];
          }
        }
      }
    }
  }
#undef A17
#undef B17
#undef STAGE_A17
#undef STAGE_B17
#undef MFMA17
#undef READ_A17
#undef READ_B17
#undef KTILE17
}

}  // namespace

extern "C" int cc_gemm_v17(const void* A, const void* B, void* C, long M,
                           long N, long K, int c_is_bf16, int persist,
                           int remap, unsigned long long stream) {
  if (K % (2 * BK) != 0 || K / BK < 4) return -2;
  int nbx = (int)((N + BN - 1) / BN);
  int nby = (int)((M + BM - 1) / BM);
  int nwg = nbx * nby;
  int grid = nwg;
  if (persist) grid = nwg < 256 ? nwg : 256;
  if (persist)
    hipLaunchKernelGGL((k_gemm_v17<1>), dim3(grid), dim3(512), 0,
                       (hipStream_t)stream, (const __bf16*)A,
                       (const __bf16*)B, C, M, N, K, c_is_bf16, nbx, nwg,
                       remap);
  else
    hipLaunchKernelGGL((k_gemm_v17<0>), dim3(grid), dim3(512), 0,
                       (hipStream_t)stream, (const __bf16*)A,
                       (const __bf16*)B, C, M, N, K, c_is_bf16, nbx, nwg,
                       remap);
  return hipGetLastError() == hipSuccess ? 0 : -1;
}

// ---- V18: ablation skeleton of the v15 persistent loop ----
// Wrong-results-by-design cost decomposition (guide §5 mistake 8:
// ablate empirically).  ABLATE: 0 = full kernel; 1 = no glds staging
// and no vmcnt drains (stale LDS); 2 = no fragment ds_reads (stale
// regs); 3 = both; 4 = no barriers (races, full compute); 5 = MFMA +
// barriers only.  Only mode 0 computes C correctly.
namespace {

template <int ABLATE>
__global__ __launch_bounds__(512, 1) void k_gemm_v18(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ C, long M, long N, long K, int c_is_bf16, int nbx,
    int nwg, int do_remap) {
  __shared__ __bf16 lds[2 * (BM + BN) * BK];
#define A18(b) (lds + (b) * (BM * BK))
#define B18(b) (lds + 2 * (BM * BK) + (b) * (BN * BK))
  constexpr bool DO_STAGE = ABLATE == 0 || ABLATE == 2 || ABLATE == 4;
  constexpr bool DO_READS = ABLATE == 0 || ABLATE == 1 || ABLATE == 4;
  constexpr bool DO_BARRIER = ABLATE != 4 && ABLATE != 7;
  constexpr bool DO_PRIO = ABLATE != 6 && ABLATE != 7 && ABLATE != 8 && ABLATE != 9;
  // modes 8/9: thin the barriers to 2 / 1 per K-tile (skeleton only)
  constexpr int BAR_EVERY = ABLATE == 8 ? 2 : (ABLATE == 9 ? 4 : 1);
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int waveM = wid >> 2, waveN = wid & 3;
  const long KT = K / BK;
  const long srow = wid * 16;
  const int arow_base = waveM * WM + (lane & 15);
  const int brow_base = waveN * WN + (lane & 15);
  bf16x8 bfragT[NFR][2] = {};
  bf16x8 afrag[2][2] = {};

#define STAGE_A18(t, h)                                                     \
  if constexpr (DO_STAGE)                                                   \
  stage_half15(A, K, bm + (h) * 128 + srow, M, (t) * BK,                    \
               A18((t) & 1) + ((h) * 128 + srow) * BK, lane)
#define STAGE_B18(t, h)                                                     \
  if constexpr (DO_STAGE)                                                   \
  stage_half15(B, K, bn + (h) * 128 + srow, N, (t) * BK,                    \
               B18((t) & 1) + ((h) * 128 + srow) * BK, lane)
#define BAR18()                                                             \
  if constexpr (DO_BARRIER) __builtin_amdgcn_s_barrier()
#define BARN18(ph)                                                          \
  if constexpr (DO_BARRIER) {                                               \
    if constexpr ((ph) % BAR_EVERY == 0) __builtin_amdgcn_s_barrier();      \
  }
#define DRAIN18()                                                           \
  if constexpr (DO_STAGE) asm volatile("s_waitcnt vmcnt(0)" ::: "memory")

  const int tiles = (nwg + (int)gridDim.x - 1) / (int)gridDim.x;
  long bm = 0, bn = 0;
  {
    int orig = (int)blockIdx.x;
    if (do_remap) {
      int q = nwg >> 3, r = nwg & 7;
      int xcd = orig & 7, lid = orig >> 3;
      orig = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + lid;
    }
    bm = (long)(orig / nbx) * BM;
    bn = (long)(orig % nbx) * BN;
  }
  STAGE_A18(0, 0);
  STAGE_A18(0, 1);
  STAGE_B18(0, 0);
  STAGE_B18(0, 1);
  STAGE_B18(1, 0);
  STAGE_B18(1, 1);

  for (int rep = 0; rep < tiles; rep++) {
    if ((int)blockIdx.x + rep * (int)gridDim.x >= nwg) break;
    f32x4 acc[MFR][NFR] = {};
    DRAIN18();
    BAR18();

#define PH18(q)                                                             \
  do {                                                                      \
    if constexpr (DO_READS)                                                 \
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");                    \
    if constexpr (DO_PRIO) __builtin_amdgcn_s_setprio(1);                                          \
    _Pragma("unroll") for (int g = 0; g < 2; g++) {                         \
      _Pragma("unroll") for (int m = 0; m < 2; m++) {                       \
        _Pragma("unroll") for (int n = 0; n < NFR; n++) {                   \
          acc[2 * (q) + m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(    \
              afrag[m][g], bfragT[n][g], acc[2 * (q) + m][n], 0, 0, 0);     \
        }                                                                   \
      }                                                                     \
    }                                                                       \
    if constexpr (DO_PRIO) __builtin_amdgcn_s_setprio(0);                   \
  } while (0)

#define RA18(At, q)                                                         \
  if constexpr (DO_READS) {                                                 \
    _Pragma("unroll") for (int g = 0; g < 2; g++) {                         \
      const int k16 = (g << 2) + (lane >> 4);                               \
      _Pragma("unroll") for (int m = 0; m < 2; m++) afrag[m][g] =           \
          frag_read(At, arow_base + (q) * 32 + m * FRAG, k16);              \
    }                                                                       \
  }

#define RB18(Bt)                                                            \
  if constexpr (DO_READS) {                                                 \
    _Pragma("unroll") for (int g = 0; g < 2; g++) {                         \
      const int k16 = (g << 2) + (lane >> 4);                               \
      _Pragma("unroll") for (int n = 0; n < NFR; n++) bfragT[n][g] =        \
          frag_read(Bt, brow_base + n * FRAG, k16);                         \
    }                                                                       \
  }

#define KT18(t)                                                             \
  do {                                                                      \
    const __bf16* At = A18((t) & 1);                                        \
    const __bf16* Bt = B18((t) & 1);                                        \
    RB18(Bt);                                                               \
    RA18(At, 0);                                                            \
    if ((t) + 1 < KT) STAGE_A18((t) + 1, 0);                                \
    BARN18(0);                                                              \
    PH18(0);                                                                \
    RA18(At, 1);                                                            \
    if ((t) + 1 < KT) STAGE_A18((t) + 1, 1);                                \
    BARN18(1);                                                              \
    PH18(1);                                                                \
    RA18(At, 2);                                                            \
    DRAIN18();                                                              \
    if ((t) + 2 < KT) STAGE_B18((t) + 2, 0);                                \
    BARN18(2);                                                              \
    PH18(2);                                                                \
    RA18(At, 3);                                                            \
    if ((t) + 2 < KT) STAGE_B18((t) + 2, 1);                                \
    BARN18(3);                                                              \
    PH18(3);                                                                \
  } while (0)

    for (long it = 0; it < KT / 2; ++it) {
      KT18(2 * it);
      KT18(2 * it + 1);
    }

    const long ebm = bm, ebn = bn;
    if (rep + 1 < tiles &&
        (int)blockIdx.x + (rep + 1) * (int)gridDim.x < nwg) {
      BAR18();
      int orig = (int)blockIdx.x + (rep + 1) * (int)gridDim.x;
      if (do_remap) {
        int q = nwg >> 3, r = nwg & 7;
        int xcd = orig & 7, lid = orig >> 3;
        orig = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + lid;
      }
      bm = (long)(orig / nbx) * BM;
      bn = (long)(orig % nbx) * BN;
      STAGE_A18(0, 0);
      STAGE_A18(0, 1);
      STAGE_B18(0, 0);
      STAGE_B18(0, 1);
      STAGE_B18(1, 0);
      STAGE_B18(1, 1);
    }

    const long crow_base = ebm + waveM * WM + 4 * (lane >> 4);
    const long ccol_base = ebn + waveN * WN + (lane & 15);
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
          if (c_is_bf16)
            ((unsigned short*)C)[row * N + col] = bf16_rne(acc[m][n][r]);
          else
            ((float*)C)[row * N + col] = acc[m][n][r];
        }
      }
    }
  }
#undef A18
#undef B18
#undef STAGE_A18
#undef STAGE_B18
#undef BAR18
#undef DRAIN18
#undef PH18
#undef RA18
#undef RB18
#undef KT18
}

}  // namespace

extern "C" int cc_gemm_v18(const void* A, const void* B, void* C, long M,
                           long N, long K, int c_is_bf16, int ablate,
                           int remap, unsigned long long stream) {
  if (K % (2 * BK) != 0 || K / BK < 4) return -2;
  int nbx = (int)((N + BN - 1) / BN);
  int nby = (int)((M + BM - 1) / BM);
  int nwg = nbx * nby;
  int grid = nwg < 256 ? nwg : 256;
#define L18(AB)                                                            \
  hipLaunchKernelGGL((k_gemm_v18<AB>), dim3(grid), dim3(512), 0,           \
                     (hipStream_t)stream, (const __bf16*)A,                \
                     (const __bf16*)B, C, M, N, K, c_is_bf16, nbx, nwg,    \
                     remap)
  switch (ablate) {
    case 0: L18(0); break;
    case 1: L18(1); break;
    case 2: L18(2); break;
    case 3: L18(3); break;
    case 4: L18(4); break;
    case 5: L18(5); break;
    case 6: L18(6); break;
    case 7: L18(7); break;
    case 8: L18(8); break;
    case 9: L18(9); break;
    default: return -3;
  }
#undef L18
  return hipGetLastError() == hipSuccess ? 0 : -1;
}
