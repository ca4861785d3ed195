// Fused multi-head attention for the ViT encoders — three ladder rungs
// by sequence length, all reading Q/K/V straight out of the fused-QKV
// GEMM output ([n*seq, 3*H] rows tokens) and writing O contiguous
// [n*seq, H] (the out-projection GEMM's input layout), replacing
// torch-rocm sdpa (~94 TF/s) + the qkv permute copies:
//   k_attn_small  seq <= 64   (ViT-B/32's 50): everything LDS-resident,
//                 4 WG... 2 WG/CU — see its note on the direct-global
//                 variant that measured slower here;
//   k_attn_mid    64 < seq <= 288 (ViT-L/14's 257, SigLIP-L16-256's
//                 256): Q/K fragments direct from global (16B-contiguous
//                 rows), LDS only V^T + P -> 2 WG/CU;
//   k_attn_flash  seq > 288 (SigLIP-384-class): K/V streamed in 64-row
//                 tiles with online softmax, O rescaled flash-style.
//
// Numerics (all rungs): f32 accumulation and softmax (matches sdpa's
// f32 softmax on bf16 inputs); probabilities round to bf16 before PV,
// covered by sdpa-parity tests at seq 50/100/257/288/576/700/1024 and
// the end-to-end cosine tests.

#include <hip/hip_runtime.h>

#include <vector>

#include "cc_common.hpp"
#include "cc_timing.hpp"

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

constexpr int SMAX = 64;   // padded sequence tile
constexpr int HD = 64;     // head dim
constexpr int LD = 72;     // LDS row stride (pad 8 elements vs 64 banks)

// NOTE: the direct-global Q/K variant (4 WG/CU) was tried here too and
// measured SLOWER (89 -> 126 us/launch, profiles/r01_rocprof_b32_afterfusions.txt)
// — at seq=50 the whole K tile L1/LDS-fits and re-reading it from L2
// per wave costs more than the staging saves (guide rule 7); the
// LDS-resident form below stays.  k_attn_mid (seq 257) keeps the
// direct-global form where it measured FASTER (L/14 step 80.3 -> 75.1 ms).
__global__ __launch_bounds__(256, 2) void k_attn_small(
    const __bf16* __restrict__ qkv, __bf16* __restrict__ out, long n_frames,
    int seq, int heads, int hidden, float scale) {
  // one workgroup per (frame, head)
  const long fh = blockIdx.x;
  const long frame = fh / heads;
  const int head = fh % heads;
  if (frame >= n_frames) return;

  __shared__ __bf16 lds[4 * SMAX * LD];  // Q, K, Vt, P
  __bf16* Q = lds;
  __bf16* K = lds + SMAX * LD;
  __bf16* Vt = lds + 2 * SMAX * LD;
  __bf16* P = lds + 3 * SMAX * LD;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;  // 4 waves; wave w owns S rows [16w, 16w+16)

  // ---- load Q/K/V tiles: thread t handles row t>>2, cols (t&3)*16.. ----
  {
    const int row = tid >> 2;
    const int d0 = (tid & 3) * 16;
    const long base = ((frame * seq + row) * 3) * (long)hidden + (long)head * HD;
    bf16x8 z = {};
    bf16x8 q0 = z, q1 = z, k0 = z, k1 = z, v0 = z, v1 = z;
    if (row < seq) {
      q0 = *(const bf16x8*)(qkv + base + d0);
      q1 = *(const bf16x8*)(qkv + base + d0 + 8);
      k0 = *(const bf16x8*)(qkv + base + hidden + d0);
      k1 = *(const bf16x8*)(qkv + base + hidden + d0 + 8);
      v0 = *(const bf16x8*)(qkv + base + 2 * hidden + d0);
      v1 = *(const bf16x8*)(qkv + base + 2 * hidden + d0 + 8);
    }
    *(bf16x8*)(Q + row * LD + d0) = q0;
    *(bf16x8*)(Q + row * LD + d0 + 8) = q1;
    *(bf16x8*)(K + row * LD + d0) = k0;
    *(bf16x8*)(K + row * LD + d0 + 8) = k1;
    // V transposed: Vt[d][row] = V[row][d]
#pragma unroll
    for (int e = 0; e < 8; e++) {
      Vt[(d0 + e) * LD + row] = v0[e];
      Vt[(d0 + 8 + e) * LD + row] = v1[e];
    }
  }
  __syncthreads();

  // ---- S = Q K^T * scale, rows [16w,16w+16), cols 0..63 ----
  f32x4 acc[4] = {};
  {
    const int qrow = 16 * wid + (lane & 15);
    const int k0e = 8 * (lane >> 4);
#pragma unroll
    for (int kk = 0; kk < HD; kk += 32) {
      bf16x8 qf = *(const bf16x8*)(Q + qrow * LD + kk + k0e);
#pragma unroll
      for (int n = 0; n < 4; n++) {
        bf16x8 kf = *(const bf16x8*)(K + (n * 16 + (lane & 15)) * LD + kk + k0e);
        acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf, kf, acc[n], 0, 0, 0);
      }
    }
  }

  // ---- masked, scaled softmax over cols (per row) ----
  // acc value (row = 16w + (lane>>4)*4 + reg, col = n*16 + lane&15)
  float pr[4][4];  // probabilities, same layout
  {
    const int colr = lane & 15;
#pragma unroll
    for (int reg = 0; reg < 4; reg++) {
      float mx = -1e30f;
      float sv[4];
#pragma unroll
      for (int nn = 0; nn < 4; nn++) {
        float s = acc[nn][reg] * scale;
        if (nn * 16 + colr >= seq) s = -1e30f;
        sv[nn] = s;
        mx = fmaxf(mx, s);
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off));
      float sum = 0.0f;
#pragma unroll
      for (int nn = 0; nn < 4; nn++) {
        float e = __expf(sv[nn] - mx);
        if (nn * 16 + colr >= seq) e = 0.0f;
        sv[nn] = e;
        sum += e;
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) sum += __shfl_xor(sum, off);
      const float inv = 1.0f / sum;
#pragma unroll
      for (int nn = 0; nn < 4; nn++) pr[nn][reg] = sv[nn] * inv;
    }
  }
  // write P (bf16) to LDS for the PV matmul
  {
    const int colr = lane & 15;
    const int rbase = 16 * wid + 4 * (lane >> 4);
#pragma unroll
    for (int nn = 0; nn < 4; nn++)
#pragma unroll
      for (int reg = 0; reg < 4; reg++)
        P[(rbase + reg) * LD + nn * 16 + colr] = (__bf16)pr[nn][reg];
  }
  __syncthreads();

  // ---- O = P Vt^T : rows [16w,16w+16) of O, cols d = 0..63 ----
  f32x4 oacc[4] = {};
  {
    const int prow = 16 * wid + (lane & 15);
    const int j0 = 8 * (lane >> 4);
#pragma unroll
    for (int kk = 0; kk < SMAX; kk += 32) {
      bf16x8 pf = *(const bf16x8*)(P + prow * LD + kk + j0);
#pragma unroll
      for (int n = 0; n < 4; n++) {
        bf16x8 vf = *(const bf16x8*)(Vt + (n * 16 + (lane & 15)) * LD + kk + j0);
        oacc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, vf, oacc[n], 0, 0, 0);
      }
    }
  }
  // ---- store O: (row = 16w + (lane>>4)*4 + reg, d = n*16 + lane&15) ----
  {
    const int dcol = lane & 15;
    const int rbase = 16 * wid + 4 * (lane >> 4);
#pragma unroll
    for (int nn = 0; nn < 4; nn++)
#pragma unroll
      for (int reg = 0; reg < 4; reg++) {
        const int row = rbase + reg;
        if (row < seq)
          out[(frame * seq + row) * (long)hidden + head * HD + nn * 16 + dcol] =
              (__bf16)oacc[nn][reg];
      }
  }
}


// ---- mid-sequence variant (64 < seq <= 288): ViT-L/14's seq=257 ----
// One workgroup per (frame, head), Q in 64-row tiles.  Q and K
// fragments are 16-byte contiguous in the fused-QKV layout, so they
// load STRAIGHT from global/L2 (no LDS staging) — only V^T (transposed
// image for the PV matmul) and P live in LDS: 74 KB -> 2 WG/CU, which
// doubles the waves/SIMD vs the LDS-everything form (measured 621 us
// -> see profiles/r01_attn_mid2; the LDS-everything form capped at
// ~204 TF with 1 wave/SIMD).
constexpr int SMID = 288;       // padded sequence (18 x 16-col frags)
constexpr int LDM = SMID + 8;   // LDS s-stride for Vt / P rows

__global__ __launch_bounds__(256, 2) void k_attn_mid(
    const __bf16* __restrict__ qkv, __bf16* __restrict__ out, long n_frames,
    int seq, int heads, int hidden, float scale) {
  const long fh = blockIdx.x;
  const long frame = fh / heads;
  const int head = fh % heads;
  if (frame >= n_frames) return;

  __shared__ __bf16 lds[2 * 64 * LDM];
  __bf16* Vt = lds;              // [64][LDM] rows d, cols s
  __bf16* P = lds + 64 * LDM;    // [64][LDM] rows q, cols s

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const long qkv_base = (frame * (long)seq * 3) * hidden + (long)head * HD;

  // ---- V^T load: 64 rows per pass, thread t -> row t>>2, d (t&3)*16
  {
    const int row = tid >> 2;
    const int d0 = (tid & 3) * 16;
#pragma unroll
    for (int j = 0; j < SMID / 64 + 1; j++) {
      const int r = j * 64 + row;
      if (r >= SMID) break;
      bf16x8 z = {};
      bf16x8 v0 = z, v1 = z;
      if (r < seq) {
        const long base = qkv_base + (long)r * 3 * hidden + 2 * hidden;
        v0 = *(const bf16x8*)(qkv + base + d0);
        v1 = *(const bf16x8*)(qkv + base + d0 + 8);
      }
#pragma unroll
      for (int e = 0; e < 8; e++) {
        Vt[(d0 + e) * LDM + r] = v0[e];
        Vt[(d0 + 8 + e) * LDM + r] = v1[e];
      }
    }
  }

  const int n_qtiles = (seq + 63) / 64;
  for (int qt = 0; qt < n_qtiles; qt++) {
    const int q0 = qt * 64;
    __syncthreads();  // Vt ready (first iter); P free (later iters)

    // ---- S = Q K^T: wave rows [16*wid, +16), 18 col frags; Q/K frags
    // read directly from global (16B contiguous), rows clamped to seq-1
    // (masked in softmax anyway).
    f32x4 acc[SMID / 16];
#pragma unroll
    for (int n = 0; n < SMID / 16; n++) acc[n] = f32x4{};
    {
      int qrow = q0 + 16 * wid + (lane & 15);
      qrow = qrow < seq ? qrow : seq - 1;
      const int k0e = 8 * (lane >> 4);
#pragma unroll
      for (int kk = 0; kk < HD; kk += 32) {
        bf16x8 qf =
            *(const bf16x8*)(qkv + qkv_base + (long)qrow * 3 * hidden + kk + k0e);
#pragma unroll
        for (int n = 0; n < SMID / 16; n++) {
          int krow = n * 16 + (lane & 15);
          krow = krow < seq ? krow : seq - 1;
          bf16x8 kf = *(const bf16x8*)(qkv + qkv_base + (long)krow * 3 * hidden +
                                       hidden + kk + k0e);
          acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf, kf, acc[n], 0, 0, 0);
        }
      }
    }

    // ---- softmax per row (cols masked at seq)
    {
      const int colr = lane & 15;
#pragma unroll
      for (int reg = 0; reg < 4; reg++) {
        float mx = -1e30f;
        float sv[SMID / 16];
#pragma unroll
        for (int nn = 0; nn < SMID / 16; nn++) {
          float sc = acc[nn][reg] * scale;
          if (nn * 16 + colr >= seq) sc = -1e30f;
          sv[nn] = sc;
          mx = fmaxf(mx, sc);
        }
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
          mx = fmaxf(mx, __shfl_xor(mx, off));
        float sum = 0.0f;
#pragma unroll
        for (int nn = 0; nn < SMID / 16; nn++) {
          float e = __expf(sv[nn] - mx);
          if (nn * 16 + colr >= seq) e = 0.0f;
          sv[nn] = e;
          sum += e;
        }
#pragma unroll
        for (int off = 1; off < 16; off <<= 1) sum += __shfl_xor(sum, off);
        const float inv = 1.0f / sum;
#pragma unroll
        for (int nn = 0; nn < SMID / 16; nn++) acc[nn][reg] = sv[nn] * inv;
      }
    }
    // ---- P to LDS (bf16)
    {
      const int colr = lane & 15;
      const int rbase = 16 * wid + 4 * (lane >> 4);
#pragma unroll
      for (int nn = 0; nn < SMID / 16; nn++)
#pragma unroll
        for (int reg = 0; reg < 4; reg++)
          P[(rbase + reg) * LDM + nn * 16 + colr] = (__bf16)acc[nn][reg];
    }
    __syncthreads();

    // ---- O = P V : rows [16*wid,+16), d cols 0..63
    f32x4 oacc[4] = {};
    {
      const int prow = 16 * wid + (lane & 15);
      const int j0 = 8 * (lane >> 4);
#pragma unroll
      for (int kk = 0; kk < SMID; kk += 32) {
        bf16x8 pf = *(const bf16x8*)(P + prow * LDM + kk + j0);
#pragma unroll
        for (int n = 0; n < 4; n++) {
          bf16x8 vf =
              *(const bf16x8*)(Vt + (n * 16 + (lane & 15)) * LDM + kk + j0);
          oacc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, vf, oacc[n], 0, 0, 0);
        }
      }
    }
    {
      const int dcol = lane & 15;
      const int rbase = 16 * wid + 4 * (lane >> 4);
#pragma unroll
      for (int nn = 0; nn < 4; nn++)
#pragma unroll
        for (int reg = 0; reg < 4; reg++) {
          const int row = q0 + rbase + reg;
          if (row < seq)
            out[(frame * seq + row) * (long)hidden + head * HD + nn * 16 +
                dcol] = (__bf16)oacc[nn][reg];
        }
    }
  }
}


// ---- flash variant (seq > 288): K/V streamed in 64-row tiles with
// online softmax ----
// One workgroup per (frame, head, q-tile of 64 rows); Q and K fragments
// load straight from global (16B-contiguous in the fused-QKV layout,
// same trick as k_attn_mid); per KV-tile, V^T and the probability tile
// stage through a small LDS image (~19 KB -> high occupancy).  f32
// running max/sum per row, O accumulators rescaled flash-style,
// probabilities rounded to bf16 before PV (same numerics contract as
// the resident kernels).
constexpr int FKV = 64;        // kv tile rows
constexpr int LDF = FKV + 8;   // LDS s-stride

__global__ __launch_bounds__(256, 2) void k_attn_flash(
    const __bf16* __restrict__ qkv, __bf16* __restrict__ out, long n_frames,
    int seq, int heads, int hidden, float scale) {
  const int n_qtiles = (seq + 63) / 64;
  const long fhq = blockIdx.x;
  const int qt = (int)(fhq % n_qtiles);
  const long fh = fhq / n_qtiles;
  const long frame = fh / heads;
  const int head = fh % heads;
  if (frame >= n_frames) return;

  __shared__ __bf16 lds[2 * 64 * LDF];
  __bf16* Vt = lds;              // [64 d][LDF s]
  __bf16* P = lds + 64 * LDF;    // [64 q][LDF s]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const long qkv_base = (frame * (long)seq * 3) * hidden + (long)head * HD;
  const int q0 = qt * 64;

  // per-lane row state (rows rbase..rbase+3 of this wave's 16)
  const int rbase_l = 4 * (lane >> 4);
  float m_run[4], l_run[4];
  f32x4 oacc[4] = {};
#pragma unroll
  for (int r = 0; r < 4; r++) {
    m_run[r] = -1e30f;
    l_run[r] = 0.0f;
  }

  const int n_kv = (seq + FKV - 1) / FKV;
  for (int kt = 0; kt < n_kv; kt++) {
    const int k0 = kt * FKV;
    // ---- V^T tile into LDS (thread t -> row t>>2, d (t&3)*16)
    __syncthreads();  // P/Vt free from the previous tile
    {
      const int row = tid >> 2;
      const int d0 = (tid & 3) * 16;
      bf16x8 z = {};
      bf16x8 v0 = z, v1 = z;
      if (k0 + row < seq) {
        const long base = qkv_base + (long)(k0 + row) * 3 * hidden + 2 * hidden;
        v0 = *(const bf16x8*)(qkv + base + d0);
        v1 = *(const bf16x8*)(qkv + base + d0 + 8);
      }
#pragma unroll
      for (int e = 0; e < 8; e++) {
        Vt[(d0 + e) * LDF + row] = v0[e];
        Vt[(d0 + 8 + e) * LDF + row] = v1[e];
      }
    }

    // ---- S tile = Q K^T (wave rows, 4 col frags), direct global Q/K
    f32x4 acc[4] = {};
    {
      int qrow = q0 + 16 * wid + (lane & 15);
      qrow = qrow < seq ? qrow : seq - 1;
      const int k0e = 8 * (lane >> 4);
#pragma unroll
      for (int kk = 0; kk < HD; kk += 32) {
        bf16x8 qf =
            *(const bf16x8*)(qkv + qkv_base + (long)qrow * 3 * hidden + kk + k0e);
#pragma unroll
        for (int n = 0; n < 4; n++) {
          int krow = k0 + n * 16 + (lane & 15);
          krow = krow < seq ? krow : seq - 1;
          bf16x8 kf = *(const bf16x8*)(qkv + qkv_base + (long)krow * 3 * hidden +
                                       hidden + kk + k0e);
          acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf, kf, acc[n], 0, 0, 0);
        }
      }
    }

    // ---- online softmax update (per row)
    float alpha[4];
    {
      const int colr = lane & 15;
#pragma unroll
      for (int reg = 0; reg < 4; reg++) {
        float mx = m_run[reg];
        float sv[4];
#pragma unroll
        for (int nn = 0; nn < 4; nn++) {
          float sc = acc[nn][reg] * scale;
          if (k0 + nn * 16 + colr >= seq) sc = -1e30f;
          sv[nn] = sc;
          mx = fmaxf(mx, sc);
        }
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
          mx = fmaxf(mx, __shfl_xor(mx, off));
        alpha[reg] = __expf(m_run[reg] - mx);
        float sum = 0.0f;
#pragma unroll
        for (int nn = 0; nn < 4; nn++) {
          float e = __expf(sv[nn] - mx);
          if (k0 + nn * 16 + colr >= seq) e = 0.0f;
          sv[nn] = e;
          sum += e;
        }
#pragma unroll
        for (int off = 1; off < 16; off <<= 1) sum += __shfl_xor(sum, off);
        l_run[reg] = l_run[reg] * alpha[reg] + sum;
        m_run[reg] = mx;
#pragma unroll
        for (int nn = 0; nn < 4; nn++) acc[nn][reg] = sv[nn];
      }
    }
    // ---- P tile to LDS (unnormalized probabilities, bf16)
    {
      const int colr = lane & 15;
      const int rb = 16 * wid + rbase_l;
#pragma unroll
      for (int nn = 0; nn < 4; nn++)
#pragma unroll
        for (int reg = 0; reg < 4; reg++)
          P[(rb + reg) * LDF + nn * 16 + colr] = (__bf16)acc[nn][reg];
    }
    __syncthreads();

    // ---- O = alpha * O + P V
    f32x4 pv[4] = {};
    {
      const int prow = 16 * wid + (lane & 15);
      const int j0 = 8 * (lane >> 4);
#pragma unroll
      for (int kk = 0; kk < FKV; kk += 32) {
        bf16x8 pf = *(const bf16x8*)(P + prow * LDF + kk + j0);
#pragma unroll
        for (int n = 0; n < 4; n++) {
          bf16x8 vf =
              *(const bf16x8*)(Vt + (n * 16 + (lane & 15)) * LDF + kk + j0);
          pv[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, vf, pv[n], 0, 0, 0);
        }
      }
    }
#pragma unroll
    for (int n = 0; n < 4; n++)
#pragma unroll
      for (int reg = 0; reg < 4; reg++)
        oacc[n][reg] = oacc[n][reg] * alpha[reg] + pv[n][reg];
  }

  // ---- store O / l
  {
    const int dcol = lane & 15;
    const int rb = 16 * wid + rbase_l;
#pragma unroll
    for (int nn = 0; nn < 4; nn++)
#pragma unroll
      for (int reg = 0; reg < 4; reg++) {
        const int row = q0 + rb + reg;
        if (row < seq)
          out[(frame * seq + row) * (long)hidden + head * HD + nn * 16 + dcol] =
              (__bf16)(oacc[nn][reg] / l_run[reg]);
      }
  }
}

}  // namespace

extern "C" int cc_attn_small(const void* qkv, void* out, int64_t n_frames,
                             int seq, int heads, int hidden, float scale,
                             uint64_t stream) {
  if (!qkv || !out || n_frames <= 0 || seq <= 0)
    return cc::set_error(CC_ERR_INVALID, "bad attn args");
  if (seq > SMAX || hidden != heads * HD)
    return cc::set_error(CC_ERR_UNSUPPORTED,
                         "cc_attn_small needs seq <= 64 and hd == 64");
  dim3 block(256), grid((unsigned)(n_frames * heads));
  hipEvent_t ev0, ev1;
  bool timed = cc::timed_begin(stream, &ev0, &ev1);
  hipLaunchKernelGGL(k_attn_small, grid, block, 0, (hipStream_t)stream,
                     (const __bf16*)qkv, (__bf16*)out, (long)n_frames, seq,
                     heads, hidden, scale);
  hipError_t e = hipGetLastError();
  if (timed) cc::timed_end("attn_small", stream, ev0, ev1);
  if (e != hipSuccess)
    return cc::set_error(CC_ERR_HIP, "attn launch: %s", hipGetErrorString(e));
  return CC_OK;
}

extern "C" int cc_attn_mid(const void* qkv, void* out, int64_t n_frames,
                           int seq, int heads, int hidden, float scale,
                           uint64_t stream) {
  if (!qkv || !out || n_frames <= 0 || seq <= 0)
    return cc::set_error(CC_ERR_INVALID, "bad attn args");
  if (seq <= SMAX || seq > SMID || hidden != heads * HD)
    return cc::set_error(CC_ERR_UNSUPPORTED,
                         "cc_attn_mid needs 64 < seq <= 288 and hd == 64");
  dim3 block(256), grid((unsigned)(n_frames * heads));
  hipEvent_t ev0, ev1;
  bool timed = cc::timed_begin(stream, &ev0, &ev1);
  hipLaunchKernelGGL(k_attn_mid, grid, block, 0, (hipStream_t)stream,
                     (const __bf16*)qkv, (__bf16*)out, (long)n_frames, seq,
                     heads, hidden, scale);
  hipError_t e = hipGetLastError();
  if (timed) cc::timed_end("attn_mid", stream, ev0, ev1);
  if (e != hipSuccess)
    return cc::set_error(CC_ERR_HIP, "attn_mid launch: %s", hipGetErrorString(e));
  return CC_OK;
}

extern "C" int cc_attn_flash(const void* qkv, void* out, int64_t n_frames,
                             int seq, int heads, int hidden, float scale,
                             uint64_t stream) {
  if (!qkv || !out || n_frames <= 0 || seq <= 0)
    return cc::set_error(CC_ERR_INVALID, "bad attn args");
  if (seq <= SMID || hidden != heads * HD)
    return cc::set_error(CC_ERR_UNSUPPORTED,
                         "cc_attn_flash is for seq > 288 with hd == 64 "
                         "(use attn_small/attn_mid below that)");
  const int n_qtiles = (seq + 63) / 64;
  dim3 block(256), grid((unsigned)(n_frames * heads * (long)n_qtiles));
  hipEvent_t ev0, ev1;
  bool timed = cc::timed_begin(stream, &ev0, &ev1);
  hipLaunchKernelGGL(k_attn_flash, grid, block, 0, (hipStream_t)stream,
                     (const __bf16*)qkv, (__bf16*)out, (long)n_frames, seq,
                     heads, hidden, scale);
  hipError_t e = hipGetLastError();
  if (timed) cc::timed_end("attn_flash", stream, ev0, ev1);
  if (e != hipSuccess)
    return cc::set_error(CC_ERR_HIP, "attn_flash launch: %s", hipGetErrorString(e));
  return CC_OK;
}
