// bf16 MFMA GEMM for the ViT forward — hand-written for gfx950 (CDNA4).
//
// Replaces the cuBLAS GEMMs under CLIPModel.get_image_features
// (/root/reference/cosmos_curate/models/clip.py:71): patch-embed-as-GEMM,
// QKV / attention-out projections, MLP fc1/fc2, visual projection
// (SURVEY.md §2b row 8).
//
// C[M,N] = act(A[M,K] x B[N,K]^T + bias[N]) [+ residual[M,N]];
// A,B bf16 row-major, f32 accumulate on v_mfma_f32_16x16x32_bf16, output
// f32 or bf16.  The fused epilogue (quick-gelu, residual add) removes the
// bandwidth-bound elementwise kernels between GEMMs (one 206 MB round
// trip per MLP layer at the bench shape).
//
// Design (cdna_hip_programming.md §5, step-3 ladder + measured fixes):
//   - 128x128 block tile, BK=64, 256 threads = 4 waves (2x2), each wave a
//     64x64 sub-tile = 4x4 MFMA fragments of 16x16.
//   - global->LDS via __builtin_amdgcn_global_load_lds width 16,
//     double-buffered, one barrier + vmcnt(0) per K-tile.
//   - LDS image XOR-swizzled: the 16-byte slot for (row, k16) lives at
//     k16 ^ (row & 7).  glds demands a lane-linear LDS destination, so the
//     swizzle is applied to the per-lane GLOBAL source address (guide §5
//     "pre-swizzling the per-lane global address"); fragment ds_read_b128
//     applies the same XOR.  Measured motivation: linear layout put 4
//     same-bank rows in each 16-lane read group (SQ_LDS_BANK_CONFLICT =
//     0.75x extra LDS cycles, profiles/r01_pmc).
//   - XCD-aware bijective block remap (L2 affinity, guide §5.4): block b
//     of nwg runs on XCD b%8, so adjacent output tiles share an XCD's L2.
//   - M/N edges by clamped global loads + predicated stores; K % 64 == 0
//     required (all ViT shapes comply).

#include <hip/hip_runtime.h>

#include <cstdlib>
#include <vector>

#include "cc_common.hpp"
#include "cc_timing.hpp"

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
typedef __attribute__((ext_vector_type(16))) float f32x16;

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int WM = 64, WN = 64;  // per-wave sub-tile (2x2 wave grid)
constexpr int FRAG = 16;         // mfma 16x16x32
constexpr int MFR = WM / FRAG;   // 4
constexpr int NFR = WN / FRAG;   // 4

__device__ __forceinline__ unsigned short f32_to_bf16_rne(float v) {
  union {
    float f;
    unsigned int u;
  } cv{v};
  unsigned int lsb = (cv.u >> 16) & 1;
  return (unsigned short)((cv.u + 0x7fffu + lsb) >> 16);
}

// fused activations (quick-gelu / SigLIP tanh-gelu; see epilogue notes)
template <int ACT>
__device__ __forceinline__ float cc_act(float v) {
  if constexpr (ACT == 1)
    return v * __builtin_amdgcn_rcpf(1.0f + __expf(-1.702f * v));
  if constexpr (ACT == 2) {
    // tanh(z) = 1 - 2/(e^{2z}+1): __expf+rcp beats libm tanhf
    float z = 0.7978845608028654f * (v + 0.044715f * v * v * v);
    float t = 1.0f - 2.0f * __builtin_amdgcn_rcpf(__expf(2.0f * z) + 1.0f);
    return 0.5f * v * (1.0f + t);
  }
  return v;
}

// stage one 32-row slice of a [rows x BK] bf16 tile into LDS via glds,
// with the k16 ^ (row&7) source swizzle.  lds_base = this wave's slice.
__device__ __forceinline__ void stage_slice(const __bf16* __restrict__ src,
                                            long ld, long row0, long row_limit,
                                            long k0, __bf16* lds_base,
                                            int lane) {
  const int lrow8 = lane >> 3;            // local row within each 8-row chunk
  const int slot = lane & 7;              // LDS 16B slot this lane fills
  const int gk16 = slot ^ lrow8;          // swizzle: global k16 feeding slot
#pragma unroll
  for (int j = 0; j < 4; j++) {  // 4 x 1KB chunks = 32 rows
    long grow = row0 + j * 8 + lrow8;
    grow = grow < 0 ? 0 : (grow >= row_limit ? row_limit - 1 : grow);
    const __bf16* gptr = src + grow * ld + k0 + (long)gk16 * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)gptr,
        (__attribute__((address_space(3))) unsigned int*)(lds_base + j * 8 * BK),
        16, 0, 0);
  }
}

// fragment read honoring the swizzle: elements (row, kk+8*(lane>>4)..+8)
__device__ __forceinline__ bf16x8 frag_read(const __bf16* tile, int row,
                                            int k16) {
  int slot = k16 ^ (row & 7);
  return *(const bf16x8*)(tile + (long)row * BK + slot * 8);
}

template <int ACT, bool HAS_BIAS, bool HAS_RES>  // ACT 1 = quick-gelu
__global__ __launch_bounds__(256, 2) void k_gemm_bf16(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ C, const float* __restrict__ bias,
    const __bf16* __restrict__ residual, long M, long N, long K,
    int c_is_bf16, int nbx, int nwg, int do_remap) {
  __shared__ __bf16 lds[2 * (BM + BN) * BK];  // one __shared__ object (G16 4a)
#define AS(b) (lds + (b) * (BM * BK))
#define BS(b) (lds + 2 * (BM * BK) + (b) * (BN * BK))

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int waveM = wid >> 1, waveN = wid & 1;

  // XCD-aware bijective remap of the linear block id (guide §5.4):
  // b ran on XCD b%8; give XCD x the contiguous tile range so its L2
  // sees adjacent tiles.  q = nwg/8, r = nwg%8.
  int orig = blockIdx.x;
  if (do_remap) {
    int q = nwg >> 3, r = nwg & 7;
    int xcd = orig & 7, lid = orig >> 3;
    orig = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + lid;
  }
  const long bm = (long)(orig / nbx) * BM;
  const long bn = (long)(orig % nbx) * BN;

  const long arow0 = bm + 32 * wid;
  const long brow0 = bn + 32 * wid;

  f32x4 acc[MFR][NFR] = {};

  const long KT = K / BK;
  stage_slice(A, K, arow0, M, 0, AS(0) + 32 * wid * BK, lane);
  stage_slice(B, K, brow0, N, 0, BS(0) + 32 * wid * BK, lane);

  const int arow_frag = waveM * WM + (lane & 15);
  const int brow_frag = waveN * WN + (lane & 15);
  int buf = 0;
  for (long kt = 0; kt < KT; ++kt) {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    const __bf16* At = AS(buf);
    const __bf16* Bt = BS(buf);
    // Register-double-buffered fragments (graduated v9, profiles/
    // r01_gemm_v9.log): read BOTH k-groups up front (16 ds_read_b128 into
    // 64 live VGPRs) and issue the glds m0 chain under their latency, so
    // the phase runs ONE wait + 32 back-to-back MFMAs.  The previous
    // per-group form made hipcc reuse 24 staging VGPRs = 4 full
    // lgkmcnt(0) drains per phase (+12-18% measured from this change).
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
    if (kt + 1 < KT) {
      const long k0 = (kt + 1) * BK;
      stage_slice(A, K, arow0, M, k0, AS(buf ^ 1) + 32 * wid * BK, lane);
      stage_slice(B, K, brow0, N, k0, BS(buf ^ 1) + 32 * wid * BK, lane);
    }
#pragma unroll
    for (int g = 0; g < 2; g++)
#pragma unroll
      for (int m = 0; m < MFR; m++)
#pragma unroll
        for (int n = 0; n < NFR; n++)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[g][m], bfrag[g][n], acc[m][n], 0, 0, 0);
    buf ^= 1;
  }

  // epilogue: C/D map for 16x16x32: col = lane&15, row = (lane>>4)*4 + reg.
  // Interior tiles take the unguarded path: per-element bounds branches
  // around the residual/bias loads force hipcc into one dependent
  // load+vmcnt(0) per element (guide §5 trap (c); 458 vs 632 TF measured).
  const long crow_base = bm + waveM * WM + 4 * (lane >> 4);
  const long ccol_base = bn + waveN * WN + (lane & 15);
  const bool interior = (bm + BM <= M) && (bn + BN <= N);
  if (interior) {
#pragma unroll
    for (int m = 0; m < MFR; m++) {
#pragma unroll
      for (int n = 0; n < NFR; n++) {
        const long col = ccol_base + n * FRAG;
        float bval = 0.0f;
        if constexpr (HAS_BIAS) bval = bias[col];
        // batch the residual loads (independent, one wait) before the
        // store sequence — an interleaved load/store chain serializes
        float rv[4];
        if constexpr (HAS_RES) {
#pragma unroll
          for (int r = 0; r < 4; r++)
            rv[r] = (float)residual[(crow_base + m * FRAG + r) * N + col];
        }
#pragma unroll
        for (int r = 0; r < 4; r++) {
          const long row = crow_base + m * FRAG + r;
          float v = acc[m][n][r] + bval;
          if (ACT == 1) v = v * __builtin_amdgcn_rcpf(1.0f + __expf(-1.702f * v));
          if (ACT == 2) {  // tanh-gelu (SigLIP gelu_pytorch_tanh)
            // tanh(z) = 1 - 2/(e^{2z}+1): __expf+rcp beats libm tanhf
            // (~80 us/launch at the SigLIP fc1 shape, r01_siglip_prof)
            float z = 0.7978845608028654f * (v + 0.044715f * v * v * v);
            float t = 1.0f - 2.0f * __builtin_amdgcn_rcpf(__expf(2.0f * z) + 1.0f);
            v = 0.5f * v * (1.0f + t);
          }
          if constexpr (HAS_RES) v += rv[r];
          if (c_is_bf16)
            ((unsigned short*)C)[row * N + col] = f32_to_bf16_rne(v);
          else
            ((float*)C)[row * N + col] = v;
        }
      }
    }
    return;
  }
#pragma unroll
  for (int m = 0; m < MFR; m++) {
#pragma unroll
    for (int n = 0; n < NFR; n++) {
      const long col = ccol_base + n * FRAG;
      if (col >= N) continue;
      float bval = 0.0f;
      if constexpr (HAS_BIAS) bval = bias[col];
#pragma unroll
      for (int r = 0; r < 4; r++) {
        const long row = crow_base + m * FRAG + r;
        if (row >= M) continue;
        float v = acc[m][n][r] + bval;
        if (ACT == 1) v = v * __builtin_amdgcn_rcpf(1.0f + __expf(-1.702f * v));  // quick-gelu
        if (ACT == 2) {  // tanh-gelu (SigLIP gelu_pytorch_tanh)
          // tanh(z) = 1 - 2/(e^{2z}+1): __expf+rcp beats libm tanhf
          // (~80 us/launch at the SigLIP fc1 shape, r01_siglip_prof)
          float z = 0.7978845608028654f * (v + 0.044715f * v * v * v);
          float t = 1.0f - 2.0f * __builtin_amdgcn_rcpf(__expf(2.0f * z) + 1.0f);
          v = 0.5f * v * (1.0f + t);
        }
        if constexpr (HAS_RES) v += (float)residual[row * N + col];
        if (c_is_bf16)
          ((unsigned short*)C)[row * N + col] = f32_to_bf16_rne(v);
        else
          ((float*)C)[row * N + col] = v;
      }
    }
  }
}

// 32x32x16-fragment variant: same staging/swizzle/XCD remap, wave tile
// 64x64 as 2x2 fragments of 32x32.  Measured (profiles/r01_gemm_variants):
// +19-36% over the 16x16x32 body on the K=768 ViT shapes and +7% on big
// squares (fewer, denser MFMAs: 2382 vs 2075 TF ceiling); the 16x16 body
// keeps a small edge at K >= 2048 (patch-embed), so the launcher picks by K.
template <int ACT, bool HAS_BIAS, bool HAS_RES>
__global__ __launch_bounds__(256, 2) void k_gemm_bf16_w32(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ C, const float* __restrict__ bias,
    const __bf16* __restrict__ residual, long M, long N, long K,
    int c_is_bf16, int nbx, int nwg, int do_remap) {
  __shared__ __bf16 lds[2 * (BM + BN) * BK];
#define AS32(b) (lds + (b) * (BM * BK))
#define BS32(b) (lds + 2 * (BM * BK) + (b) * (BN * BK))
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int waveM = wid >> 1, waveN = wid & 1;

  int orig = blockIdx.x;
  if (do_remap) {
    int q = nwg >> 3, r = nwg & 7;
    int xcd = orig & 7, lid = orig >> 3;
    orig = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + lid;
  }
  const long bm = (long)(orig / nbx) * BM;
  const long bn = (long)(orig % nbx) * BN;
  const long arow0 = bm + 32 * wid;
  const long brow0 = bn + 32 * wid;

  f32x16 acc[2][2] = {};
  const long KT = K / BK;
  stage_slice(A, K, arow0, M, 0, AS32(0) + 32 * wid * BK, lane);
  stage_slice(B, K, brow0, N, 0, BS32(0) + 32 * wid * BK, lane);
  const int arow = waveM * WM + (lane & 31);
  const int brow = waveN * WN + (lane & 31);
  int buf = 0;
  // NOTE: the v9 register-double-buffer restructure (16x16 body above)
  // was tried here too and measured SLOWER (qkv 570 vs 582 TF,
  // profiles/r01_v9_graduated.log) — the w32 body keeps the per-k-step
  // form, and the default dispatch no longer selects it (see launcher).
  for (long kt = 0; kt < KT; ++kt) {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    if (kt + 1 < KT) {
      const long k0 = (kt + 1) * BK;
      stage_slice(A, K, arow0, M, k0, AS32(buf ^ 1) + 32 * wid * BK, lane);
      stage_slice(B, K, brow0, N, k0, BS32(buf ^ 1) + 32 * wid * BK, lane);
    }
    const __bf16* At = AS32(buf);
    const __bf16* Bt = BS32(buf);
#pragma unroll
    for (int kk = 0; kk < BK; kk += 16) {
      // 32x32x16 A/B fragment: lane l holds row l&31, k 8*(l>>5)..+8
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
  // C/D map 32x32x16: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
  const long col0 = bn + waveN * WN + (lane & 31);
  const long row0 = bm + waveM * WM + 4 * (lane >> 5);
  const bool interior = (bm + BM <= M) && (bn + BN <= N);
  if (interior) {
#pragma unroll
    for (int m = 0; m < 2; m++) {
#pragma unroll
      for (int n = 0; n < 2; n++) {
        const long col = col0 + n * 32;
        float bval = 0.0f;
        if constexpr (HAS_BIAS) bval = bias[col];
        float rv[16];
        if constexpr (HAS_RES) {
#pragma unroll
          for (int reg = 0; reg < 16; reg++)
            rv[reg] = (float)residual[
                (row0 + m * 32 + (reg & 3) + 8 * (reg >> 2)) * N + col];
        }
#pragma unroll
        for (int reg = 0; reg < 16; reg++) {
          const long row = row0 + m * 32 + (reg & 3) + 8 * (reg >> 2);
          float v = acc[m][n][reg] + bval;
          if (ACT == 1) v = v * __builtin_amdgcn_rcpf(1.0f + __expf(-1.702f * v));
          if (ACT == 2) {  // tanh-gelu (SigLIP gelu_pytorch_tanh)
            // tanh(z) = 1 - 2/(e^{2z}+1): __expf+rcp beats libm tanhf
            // (~80 us/launch at the SigLIP fc1 shape, r01_siglip_prof)
            float z = 0.7978845608028654f * (v + 0.044715f * v * v * v);
            float t = 1.0f - 2.0f * __builtin_amdgcn_rcpf(__expf(2.0f * z) + 1.0f);
            v = 0.5f * v * (1.0f + t);
          }
          if constexpr (HAS_RES) v += rv[reg];
          if (c_is_bf16)
            ((unsigned short*)C)[row * N + col] = f32_to_bf16_rne(v);
          else
            ((float*)C)[row * N + col] = v;
        }
      }
    }
    return;
  }
#pragma unroll
  for (int m = 0; m < 2; m++) {
#pragma unroll
    for (int n = 0; n < 2; n++) {
      const long col = col0 + n * 32;
      if (col >= N) continue;
      float bval = 0.0f;
      if constexpr (HAS_BIAS) bval = bias[col];
#pragma unroll
      for (int reg = 0; reg < 16; reg++) {
        const long row = row0 + m * 32 + (reg & 3) + 8 * (reg >> 2);
        if (row >= M) continue;
        float v = acc[m][n][reg] + bval;
        if (ACT == 1) v = v * __builtin_amdgcn_rcpf(1.0f + __expf(-1.702f * v));
        if (ACT == 2) {  // tanh-gelu (SigLIP gelu_pytorch_tanh)
          // tanh(z) = 1 - 2/(e^{2z}+1): __expf+rcp beats libm tanhf
          // (~80 us/launch at the SigLIP fc1 shape, r01_siglip_prof)
          float z = 0.7978845608028654f * (v + 0.044715f * v * v * v);
          float t = 1.0f - 2.0f * __builtin_amdgcn_rcpf(__expf(2.0f * z) + 1.0f);
          v = 0.5f * v * (1.0f + t);
        }
        if constexpr (HAS_RES) v += (float)residual[row * N + col];
        if (c_is_bf16)
          ((unsigned short*)C)[row * N + col] = f32_to_bf16_rne(v);
        else
          ((float*)C)[row * N + col] = v;
      }
    }
  }
#undef AS32
#undef BS32
}

// ---- 256x256 8-phase template kernel (guide "The 256² 8-phase
// template", schedule derived + race-screened through tools/gemm_v10
// ... v22; round-2 ladder in DESIGN.md §9a) ----
// 512 threads = 8 waves (2M x 4N), per-wave 128x64 = 8x4 frags; LDS
// 128 KiB; PERSISTENT fleet (grid = min(nwg, 256)); raw s_barrier
// phases; COUNTED vmcnt(4) publishes at tile tops (B(t+2) spans each
// boundary in flight); T19 scheduler weave of next-phase ds_reads into
// the MFMA clusters; template-selected epilogue (LDS-staged coalesced
// for fused bias/act/residual, scalar otherwise).  Serves every shape
// whose 256-tile grid fills the fleet (the ViT GEMM mix; batch-64
// per-shape TF in profiles/r02_gemm_v21.log + the epilogue-on numbers
// in profiles/r02_epi_ab2.log).
constexpr int BM2 = 256, BN2 = 256;
constexpr int WM2 = 128, WN2 = 64;
constexpr int MFR2 = WM2 / FRAG;  // 8
constexpr int NFR2 = WN2 / FRAG;  // 4

// one wave stages its 16-row slice of a 128-row half-tile: 2 glds.
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

template <int ACT, bool HAS_BIAS, bool HAS_RES, bool EPI_STAGED>
__global__ __launch_bounds__(512, 1) void k_gemm_bf16_t256(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ C, const float* __restrict__ bias,
    const __bf16* __restrict__ residual, long M, long N, long K,
    int c_is_bf16, int nbx, int nwg, int order_mode) {
  // Round-2 structure: PERSISTENT multi-tile (grid = min(nwg, 256), one
  // WG per CU) around the round-1 8-phase glds schedule.  Each WG walks
  // tiles bid, bid+grid, ... and issues the NEXT tile's 12-glds
  // prologue BEFORE the epilogue stores, so prologue HBM latency hides
  // under the C writeback instead of stalling a fresh WG.  Measured
  // +18-27% at every batch-64 ViT shape and +13% at square8k
  // (profiles/r02_gemm_persist.log: fc1 711->874, fc2 795->968, patch
  // 814->964, qkv 724->887 TF within one box).  order_mode: 0 = plain
  // stride; 1 = stride + bijective XCD remap (skinny-N grids, L2 row
  // sharing).  Sync structure per K-tile is UNCHANGED from the
  // race-screened v11 (vmcnt(0) publish at ph3, leading barriers).
  __shared__ __bf16 lds[2 * (BM2 + BN2) * BK];  // 128 KiB
#define A2T(b) (lds + (b) * (BM2 * BK))
#define B2T(b) (lds + 2 * (BM2 * BK) + (b) * (BN2 * BK))
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int waveM = wid >> 2, waveN = wid & 3;
  const long KT = K / BK;  // launcher guarantees even, >= 4
  const long srow = wid * 16;
  const int arow_base = waveM * WM2 + (lane & 15);
  const int brow_base = waveN * WN2 + (lane & 15);
  bf16x8 bfragT[NFR2][2];  // held for the whole K-tile
  bf16x8 afrag[2][2];      // one quadrant: 2 M-frags x 2 k-steps

#define STAGE_A2(t, h)                                                      \
  stage_half(A, K, bm + (h) * 128 + srow, M, (t) * BK,                      \
             A2T((t) & 1) + ((h) * 128 + srow) * BK, lane)
#define STAGE_B2(t, h)                                                      \
  stage_half(B, K, bn + (h) * 128 + srow, N, (t) * BK,                      \
             B2T((t) & 1) + ((h) * 128 + srow) * BK, lane)

  const int tiles = (nwg + (int)gridDim.x - 1) / (int)gridDim.x;
  long bm, bn;
  {
    int orig = (int)blockIdx.x;
    if (order_mode == 1) {
      int q = nwg >> 3, r = nwg & 7;
      int xcd = orig & 7, lid = orig >> 3;
      orig = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + lid;
    }
    bm = (long)(orig / nbx) * BM2;
    bn = (long)(orig % nbx) * BN2;
  }
  // first prologue: A(0) h0,h1, B(0) h0,h1, B(1) h0,h1 (12 glds); later
  // tiles' prologues issue under the previous epilogue.
  STAGE_A2(0, 0);
  STAGE_A2(0, 1);
  STAGE_B2(0, 0);
  STAGE_B2(0, 1);
  STAGE_B2(1, 0);
  STAGE_B2(1, 1);

  for (int rep = 0; rep < tiles; rep++) {
    if ((int)blockIdx.x + rep * (int)gridDim.x >= nwg) break;
    f32x4 acc[MFR2][NFR2] = {};
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // rep-top drain: the
    __builtin_amdgcn_s_barrier();  // one full drain left (covers the 12
    // prologue glds AND the previous epilogue's in-flight stores; a
    // counted form here would need codegen-dependent store counts)

#define PHASE_MFMA2(q)                                                      \
  do {                                                                      \
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");                      \
    /* T19 dictation (round 2, +1-3%/shape, profiles/r02_gemm_v21.log):  */ \
    /* weave the next phase's ds_reads and the glds issues between the   */ \
    /* 16 MFMAs so their latency hides under the cluster (the source-    */ \
    /* level version of this, v12, regressed; the compile-time directive */ \
    /* does it without perturbing issue order elsewhere).  setprio       */ \
    /* dropped: it fenced the scheduler out of exactly this interleave.  */ \
    /* pattern 3 of the sweep (profiles/r02_t19_sweep.log): lead with   */ \
    /* 2 MFMAs, then weave [dsr, 2 MFMA, vmem] x6 — best at every ViT   */ \
    /* shape (patch 1041, qkv 916, fc1 914, fc2 1056 TF vs 875-1009     */ \
    /* for the coarser patterns).                                       */ \
    __builtin_amdgcn_sched_group_barrier(0x8 /*MFMA*/, 2, 0);               \
    _Pragma("unroll") for (int gi = 0; gi < 6; gi++) {                      \
      __builtin_amdgcn_sched_group_barrier(0x100 /*DS_READ*/, 1, 0);        \
      __builtin_amdgcn_sched_group_barrier(0x8 /*MFMA*/, 2, 0);             \
      __builtin_amdgcn_sched_group_barrier(0x10 /*VMEM*/, 1, 0);            \
    }                                                                       \
    _Pragma("unroll") for (int g = 0; g < 2; g++) {                         \
      _Pragma("unroll") for (int m = 0; m < 2; m++) {                       \
        _Pragma("unroll") for (int n = 0; n < NFR2; n++) {                  \
          acc[2 * (q) + m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(    \
              afrag[m][g], bfragT[n][g], acc[2 * (q) + m][n], 0, 0, 0);     \
        }                                                                   \
      }                                                                     \
    }                                                                       \
  } while (0)
  /* NOTE: no trailing barrier per phase (v11, profiles/r01_v11_ab.log,
     +5-8%): the B-buffer write-after-read hazard needs only 2-phase
     separation, which the per-phase leading barrier provides — wave W
     drains its reads (lgkm0) before it reaches the NEXT phase's
     barrier, and any other wave's conflicting glds issues only after
     passing that barrier, two phases later.  Race-screened +
     determinism-checked at 774-3156-block grids. */

#define READ_A2(At, q)                                                      \
  _Pragma("unroll") for (int g = 0; g < 2; g++) {                           \
    const int k16 = (g << 2) + (lane >> 4);                                 \
    _Pragma("unroll") for (int m = 0; m < 2; m++) afrag[m][g] =             \
        frag_read(At, arow_base + (q) * 32 + m * FRAG, k16);                \
  }

#define READ_B2(Bt)                                                         \
  _Pragma("unroll") for (int g = 0; g < 2; g++) {                           \
    const int k16 = (g << 2) + (lane >> 4);                                 \
    _Pragma("unroll") for (int n = 0; n < NFR2; n++) bfragT[n][g] =         \
        frag_read(Bt, brow_base + n * FRAG, k16);                           \
  }

// one K-tile = 4 phases; stages the next A halves in ph1-2 and the
// tile-after-next B halves in ph3-4 (issue order proves vmcnt(4)):
#define KTILE2(t)                                                           \
  do {                                                                      \
    const __bf16* At = A2T((t) & 1);                                        \
    const __bf16* Bt = B2T((t) & 1);                                        \
    /* counted publish for THIS tile's operands, placed as late as      */ \
    /* possible (after the previous tile's MFMA(3) instead of before    */ \
    /* it): newest-first per-wave queue here is [B(t+1):4, A(t):4, ...] */ \
    /* so vmcnt(4) retires A(t)/B(t) and keeps B(t+1) in flight.  At    */ \
    /* tile 0 of a rep the queue is already drained (rep-top vmcnt(0))  */ \
    /* and this is a no-op.                                             */ \
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");                        \
    READ_B2(Bt);                                                            \
    READ_A2(At, 0);                                                         \
    if ((t) + 1 < KT) STAGE_A2((t) + 1, 0);                                 \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA2(0);                                                         \
    READ_A2(At, 1);                                                         \
    if ((t) + 1 < KT) STAGE_A2((t) + 1, 1);                                 \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA2(1);                                                         \
    READ_A2(At, 2);                                                         \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA2(2);                                                         \
    READ_A2(At, 3);                                                         \
    /* COUNTED publish (round 2, replaces the per-tile vmcnt(0) drain):  */ \
    /* the next tile's LDS reads need A(t+1) [ph1-2, 4 loads/wave] and   */ \
    /* B(t+1) [issued one tile earlier, strictly older] landed.  After   */ \
    /* issuing B(t+2)'s 4 loads, the per-wave VMEM queue newest-first is */ \
    /* [B(t+2):4, A(t+1):4, <older>]; vmcnt(4) retires everything except */ \
    /* B(t+2), which keeps spanning the tile boundary.  r01's            */ \
    /* "nondeterministic counted wait" (profiles/r01_t256_det2.log) sat  */ \
    /* BEFORE the B(t+2) issues with exactly 4 loads outstanding — a     */ \
    /* no-op wait, not out-of-order retirement; this placement fixes the */ \
    /* count.  Verified: 6-run multi-shape determinism screen + full GPU */ \
    /* parity suite (profiles/r02_gemm_v19.log).  Measured: fc1 864->905,*/ \
    /* fc2 950->1023, patch 940->1004 TF within one box.                 */ \
    if ((t) + 2 < KT) {                                                     \
      STAGE_B2((t) + 2, 0);                                                 \
      STAGE_B2((t) + 2, 1);                                                 \
      /* publish moved to the NEXT tile's top (one more MFMA cluster of  */ \
      /* latency cover at the same count)                                */ \
    } else {                                                                \
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory"); /* tail drain */     \
    }                                                                       \
    __builtin_amdgcn_s_barrier();                                           \
    PHASE_MFMA2(3);                                                         \
  } while (0)

    for (long it = 0; it < KT / 2; ++it) {
      KTILE2(2 * it);
      KTILE2(2 * it + 1);
    }

    const long ebm = bm, ebn = bn;
    // barrier before the epilogue: the bf16 epilogue below stages C
    // through A-buf1, whose rows 96-127/224-255 other waves were still
    // reading in the last phase; the next tile's prologue (further
    // below) overwrites A-buf0/B-bufs with the same hazard.
    __builtin_amdgcn_s_barrier();
    // advance + issue the next tile's prologue BEFORE the epilogue so
    // its 12-glds HBM latency hides under the C writeback.
    if (rep + 1 < tiles &&
        (int)blockIdx.x + (rep + 1) * (int)gridDim.x < nwg) {
      int orig = (int)blockIdx.x + (rep + 1) * (int)gridDim.x;
      if (order_mode == 1) {
        int q = nwg >> 3, r = nwg & 7;
        int xcd = orig & 7, lid = orig >> 3;
        orig = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + lid;
      }
      bm = (long)(orig / nbx) * BM2;
      bn = (long)(orig % nbx) * BN2;
      STAGE_A2(0, 0);
      STAGE_A2(0, 1);
      STAGE_B2(0, 0);
      STAGE_B2(0, 1);
      STAGE_B2(1, 0);
      STAGE_B2(1, 1);
    }

    // ---- epilogue ----
    // bf16 output (all tower-internal GEMMs): LDS-staged + coalesced.
    // The fragment C/D map scatters 2-byte lanes (col = lane&15, row =
    // 4*(lane>>4)+r): storing it directly costs one 2 B store per
    // element and the same pattern for the fused residual loads — the
    // dominant cost at the thin ViT shapes once the K-loop is fed
    // (attn_out 429 vs ~790 TF bare, profiles/r02_persist_ab.log).
    // Instead: 4 passes of 2 M-fragments stage bf16(act(acc+bias))
    // through this wave's free 4 KB slice of A-buf1 (last read before
    // the barrier above; the next prologue writes only A-buf0/B-bufs),
    // then read back row-contiguous 16 B chunks, fuse the residual add
    // in f32, and store full b128s (8 lanes cover 128 contiguous
    // bytes of a C row).  Residual numerics: the pre-residual value is
    // rounded to bf16 once before the f32 add — one extra rounding vs
    // the scalar path, inside the embedding-cosine contract.
    const long crow_base = ebm + waveM * WM2 + 4 * (lane >> 4);
    const long ccol_base = ebn + waveN * WN2 + (lane & 15);
    if (c_is_bf16 && EPI_STAGED) {
      __bf16* slice = A2T(1) + wid * 2048;  // 4 KB per wave
      const long wrow0 = ebm + waveM * WM2;
      const long wcol0 = ebn + waveN * WN2;
      // bias depends only on the n-fragment: 4 loads per lane, hoisted
      // out of the 4x2 pass loop (was reloaded 8x)
      float bhoist[NFR2] = {};
      if constexpr (HAS_BIAS) {
#pragma unroll
        for (int n = 0; n < NFR2; n++) {
          const long col = ccol_base + n * FRAG;
          bhoist[n] = (col < N) ? bias[col] : 0.0f;
        }
      }
#pragma unroll
      for (int p = 0; p < 4; p++) {
#pragma unroll
        for (int mm = 0; mm < 2; mm++) {
          const int m = 2 * p + mm;
#pragma unroll
          for (int n = 0; n < NFR2; n++) {
            const float bval = bhoist[n];
            const int lrow_b = mm * 16 + 4 * (lane >> 4);
            const int lcol = n * FRAG + (lane & 15);
#pragma unroll
            for (int r = 0; r < 4; r++) {
              float v = cc_act<ACT>(acc[m][n][r] + bval);
              *(unsigned short*)(slice + (lrow_b + r) * 64 + lcol) =
                  f32_to_bf16_rne(v);
            }
          }
        }
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
        for (int i = 0; i < 4; i++) {
          const int elem = i * 512 + lane * 8;
          const int lrow = elem >> 6;          // /64
          const int lcol8 = elem & 63;         // 8-elem aligned
          const long grow = wrow0 + p * 32 + lrow;
          const long gcol = wcol0 + lcol8;
          if (grow >= M || gcol >= N) continue;  // N % 8 == 0 always
          bf16x8 v8 = *(const bf16x8*)(slice + elem);
          if constexpr (HAS_RES) {
            const bf16x8 r8 =
                *(const bf16x8*)(residual + grow * N + gcol);
#pragma unroll
            for (int e = 0; e < 8; e++)
              v8[e] = (__bf16)((float)v8[e] + (float)r8[e]);
          }
          *(bf16x8*)((__bf16*)C + grow * N + gcol) = v8;
        }
      }
    } else {
      // scalar store path (f32 outputs always; bf16 when epi_staged=0).
      // bias hoisted exactly like the staged path: 4 loads per lane
      // instead of one dependent load per fragment (the old
      // per-element-load serialization, guide trap (c)).
      const bool interior = (ebm + BM2 <= M) && (ebn + BN2 <= N);
      float bhoist2[NFR2] = {};
      if constexpr (HAS_BIAS) {
#pragma unroll
        for (int n = 0; n < NFR2; n++) {
          const long col = ccol_base + n * FRAG;
          bhoist2[n] = (interior || col < N) ? bias[col] : 0.0f;
        }
      }
#pragma unroll
      for (int m = 0; m < MFR2; m++) {
#pragma unroll
        for (int n = 0; n < NFR2; n++) {
          const long col = ccol_base + n * FRAG;
          if (!interior && col >= N) continue;
          const float bval = bhoist2[n];
#pragma unroll
          for (int r = 0; r < 4; r++) {
            const long row = crow_base + m * FRAG + r;
            if (!interior && row >= M) continue;
            float v = cc_act<ACT>(acc[m][n][r] + bval);
            if constexpr (HAS_RES) v += (float)residual[row * N + col];
            if (c_is_bf16)
              ((unsigned short*)C)[row * N + col] = f32_to_bf16_rne(v);
            else
              ((float*)C)[row * N + col] = v;
          }
        }
      }
    }
  }
#undef A2T
#undef B2T
#undef STAGE_A2
#undef STAGE_B2
#undef PHASE_MFMA2
#undef READ_A2
#undef READ_B2
#undef KTILE2
}

}  // namespace

extern "C" int cc_gemm_bf16_ex(const void* A, const void* B, void* C,
                               int64_t M, int64_t N, int64_t K,
                               const float* bias, int c_dtype, int act,
                               const void* residual, uint64_t stream) {
  if (!A || !B || !C || M <= 0 || N <= 0 || K <= 0)
    return cc::set_error(CC_ERR_INVALID, "bad gemm args");
  if (K % BK != 0)
    return cc::set_error(CC_ERR_UNSUPPORTED,
                         "cc_gemm_bf16 requires K %% 64 == 0 (got %lld)",
                         (long long)K);
  int nbx = (int)((N + BN - 1) / BN);
  int nby = (int)((M + BM - 1) / BM);
  int nwg = nbx * nby;
  dim3 block(256);
  dim3 grid(nwg);
  hipEvent_t ev0 = nullptr, ev1 = nullptr;
  bool timed = cc::timed_begin(stream, &ev0, &ev1);
  // After the v9 register-double-buffer restructure the 16x16x32 body
  // wins at EVERY measured shape (profiles/r01_v9_graduated.log: qkv
  // 671 vs 570, squares 1051/1059 vs 1015/897 TF), so it is the default
  // everywhere; the old K<2048 rule picked the 32x32x16 body.
  // CC_GEMM_WIDE=1 re-enables the w32 body for A/B experiments.
  static const int wide_env = [] {
    const char* e = getenv("CC_GEMM_WIDE");
    return e ? atoi(e) : -1;
  }();
  const bool wide = wide_env >= 0 ? wide_env != 0 : false;
  // 256² 8-phase template for the shapes its halved A/B re-read traffic
  // pays on (measured profiles/r01_v10_b64.log: wins every shape with
  // K>=1536 or N>=1536; loses only N=768 && K=768 at large M).
  // CC_GEMM_TILE=0|1 forces the 128²/256² body for A/B experiments.
  static const int tile_env = [] {
    const char* e = getenv("CC_GEMM_TILE");
    return e ? atoi(e) : -1;
  }();
  const bool t256_ok = (K % 128 == 0) && (K >= 256) && M > BM2 / 2;
  // round-2: the t256 body is persistent (grid = min(nwg, 256)); it now
  // also wins the N=768 && K=768 shape (out64 720 -> 796 TF,
  // profiles/r02_gemm_persist.log), so the only shapes kept on the 128
  // body are those whose 256-tile grid would underfill the persistent
  // fleet (nwg256 < 256, e.g. the tiny visual-projection GEMM).
  const int nbx2 = (int)((N + BN2 - 1) / BN2);
  const int nby2 = (int)((M + BM2 - 1) / BM2);
  const int nwg2 = nbx2 * nby2;
  const bool t256 =
      !wide && t256_ok &&
      (tile_env >= 0 ? tile_env != 0
                     : (K >= 1536 || N >= 1536 || nwg2 >= 256));
  const bool hb = bias != nullptr;
  const bool hr = residual != nullptr;
  // XCD remap only for outputs that spill L3 (measured negative on the
  // L2-resident ViT shapes, profiles/r01_opt3)
  int remap = ((M * N) > (48LL << 20)) ? 1 : 0;
  if (t256) {
    nbx = nbx2;
    nby = nby2;
    nwg = nwg2;
    // persistent fleet: 1 WG/CU; CC_GEMM_PERSIST=0 restores grid = nwg
    // (one tile per WG) for A/B experiments.
    static const int persist_env = [] {
      const char* e = getenv("CC_GEMM_PERSIST");
      return e ? atoi(e) : 1;
    }();
    grid = dim3(persist_env && nwg > 256 ? 256 : nwg);
    block = dim3(512);
    // stride + XCD remap order pays on skinny-N grids (row-band L2
    // sharing: qkv/fc1/fc2/patch +9-13%); on wide grids plain stride
    // wins (square8k 1295 vs 944 — profiles/r02_gemm_persist.log)
    remap = nbx <= 16 ? 1 : 0;
  }
  // bf16 epilogue: LDS-staged coalesced when the launch fuses bias/act/
  // residual (A/B: tower 582 -> 812 TF, bench 3184 -> 4188 clips/s,
  // profiles/r02_epi_ab.log), scalar for bare C=A*B^T launches where the
  // 4-pass staging is pure overhead (bare fc1 928 scalar vs 825 staged,
  // profiles/r02_gemm_v20.log).  A TEMPLATE parameter, not a runtime
  // branch: carrying both epilogues in one instantiation spilled VGPRs
  // in the rep loop and cost ~10% bare (profiles/r02_t19_sweep.log,
  // prod column vs the single-epilogue tool kernels).
  // CC_GEMM_EPI=0|1 forces it for A/B.
  static const int epi_env = [] {
    const char* e = getenv("CC_GEMM_EPI");
    return e ? atoi(e) : -1;
  }();
  const int epi_auto = (bias != nullptr || residual != nullptr || act != 0) ? 1 : 0;
  const bool epi_staged = (epi_env >= 0 ? epi_env : epi_auto) != 0;
#define CC_LAUNCH_GEMM(KER, A_, HB, HR)                                       \
  hipLaunchKernelGGL((KER<A_, HB, HR>), grid, block, 0, (hipStream_t)stream,  \
                     (const __bf16*)A, (const __bf16*)B, C, bias,             \
                     (const __bf16*)residual, (long)M, (long)N, (long)K,      \
                     c_dtype == 1 ? 1 : 0, nbx, nwg, remap)
#define CC_LAUNCH_T256(A_, HB, HR)                                            \
  do {                                                                        \
    if (epi_staged)                                                           \
      hipLaunchKernelGGL((k_gemm_bf16_t256<A_, HB, HR, true>), grid, block,   \
                         0, (hipStream_t)stream, (const __bf16*)A,            \
                         (const __bf16*)B, C, bias, (const __bf16*)residual,  \
                         (long)M, (long)N, (long)K, c_dtype == 1 ? 1 : 0,     \
                         nbx, nwg, remap);                                    \
    else                                                                      \
      hipLaunchKernelGGL((k_gemm_bf16_t256<A_, HB, HR, false>), grid, block,  \
                         0, (hipStream_t)stream, (const __bf16*)A,            \
                         (const __bf16*)B, C, bias, (const __bf16*)residual,  \
                         (long)M, (long)N, (long)K, c_dtype == 1 ? 1 : 0,     \
                         nbx, nwg, remap);                                    \
  } while (0)
#define CC_DISPATCH_ACT(KER, A_)                                              \
  do {                                                                        \
    if (hb && hr) CC_LAUNCH_GEMM(KER, A_, true, true);                        \
    else if (hb) CC_LAUNCH_GEMM(KER, A_, true, false);                        \
    else if (hr) CC_LAUNCH_GEMM(KER, A_, false, true);                        \
    else CC_LAUNCH_GEMM(KER, A_, false, false);                               \
  } while (0)
#define CC_DISPATCH(KER)                                                      \
  do {                                                                        \
    if (act == 1) CC_DISPATCH_ACT(KER, 1);                                    \
    else if (act == 2) CC_DISPATCH_ACT(KER, 2);  /* tanh-gelu (SigLIP) */     \
    else CC_DISPATCH_ACT(KER, 0);                                             \
  } while (0)
  if (wide)
    CC_DISPATCH(k_gemm_bf16_w32);
  else if (t256) {
#define CC_DISPATCH_ACT_T256(A_)                                              \
  do {                                                                        \
    if (hb && hr) CC_LAUNCH_T256(A_, true, true);                             \
    else if (hb) CC_LAUNCH_T256(A_, true, false);                             \
    else if (hr) CC_LAUNCH_T256(A_, false, true);                             \
    else CC_LAUNCH_T256(A_, false, false);                                    \
  } while (0)
    if (act == 1) CC_DISPATCH_ACT_T256(1);
    else if (act == 2) CC_DISPATCH_ACT_T256(2);
    else CC_DISPATCH_ACT_T256(0);
#undef CC_DISPATCH_ACT_T256
  } else
    CC_DISPATCH(k_gemm_bf16);
#undef CC_DISPATCH
#undef CC_DISPATCH_ACT
#undef CC_LAUNCH_GEMM
  hipError_t e = hipGetLastError();
  if (timed) cc::timed_end("gemm_bf16", stream, ev0, ev1);
  if (e != hipSuccess)
    return cc::set_error(CC_ERR_HIP, "gemm launch: %s", hipGetErrorString(e));
  return CC_OK;
}

extern "C" int cc_gemm_bf16(const void* A, const void* B, void* C, int64_t M,
                            int64_t N, int64_t K, const float* bias,
                            int c_dtype, uint64_t stream) {
  return cc_gemm_bf16_ex(A, B, C, M, N, K, bias, c_dtype, 0, nullptr, stream);
}
