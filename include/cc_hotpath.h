/* cc_hotpath.h — C ABI of the MI355X-native cosmos-curate hot path.
 *
 * One shared library (libcchot.so, built from cosmos_curate_amd/csrc/ by
 * hipcc for gfx950) exporting the functionality the reference obtains from
 * external native libraries.  Each entry point cites
 * the reference interface it replaces (file:line relative to
 * /root/reference/cosmos_curate/).  Plain pointers + sizes only; device
 * memory is referenced by raw HIP device pointers allocated either by the
 * caller's framework (torch) or by cc_malloc below; `stream` arguments are
 * hipStream_t handles passed as uint64 (0 = default stream).
 *
 * Host-side entry points (cc_demux_*) never touch the GPU and work in a
 * GPU-less container; every cc_* device function requires a visible GPU and
 * fails with CC_ERR_HIP otherwise (no CPU fallback anywhere).
 *
 * Errors: every function returns 0 on success or a negative CC_ERR_* code;
 * cc_last_error() returns a thread-local message for the last failure.
 * Thread-safety: handles are single-threaded; distinct handles independent.
 */

#ifndef CC_HOTPATH_H
#define CC_HOTPATH_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

enum {
  CC_OK = 0,
  CC_ERR_INVALID = -1,     /* bad argument */
  CC_ERR_PARSE = -2,       /* malformed container */
  CC_ERR_NOMEM = -3,
  CC_ERR_HIP = -4,         /* HIP runtime failure (incl. no GPU) */
  CC_ERR_NO_ROCDECODE = -5,/* VCN decode library not present at runtime */
  CC_ERR_UNSUPPORTED = -6,
};

const char* cc_last_error(void);

/* ---- version / capability probes ---------------------------------- */
/* Returns CC_OK when a HIP device is visible and usable. */
int cc_hip_available(void);
/* Returns CC_OK when librocdecode can be dlopened (VCN decode usable). */
int cc_rocdecode_available(void);

/* ---- host MP4 demux ------------------------------------------------
 * Replaces PyNvDemuxer (nvcodec_utils.py:224, fps at :124) and is the PTS
 * source behind get_video_timestamps (pipelines/video/utils/
 * decoder_utils.py:230-278).  Pure host C++; no codec work. */
typedef struct cc_demux cc_demux_t;

typedef struct cc_video_info {
  uint32_t width, height;
  uint32_t timescale;
  uint32_t num_samples;
  uint32_t num_sync_samples;
  int32_t  codec;           /* 0 = h264(avc1/avc3), 1 = hevc(hvc1/hev1) */
  double   duration_s;      /* track media duration in seconds */
  double   avg_fps;         /* num_samples / duration */
} cc_video_info;

int  cc_demux_open(const uint8_t* data, size_t size, cc_demux_t** out);
int  cc_demux_probe(const cc_demux_t* d, cc_video_info* info);
/* Sorted presentation timestamps in seconds, float32 — the exact
 * get_video_timestamps contract (decoder_utils.py:275-278).  `cap` is the
 * capacity of `out`; *n gets the sample count. */
int  cc_demux_timestamps(const cc_demux_t* d, float* out, size_t cap, size_t* n);
/* AnnexB packet for sample `index` (decode order): start-code NALs, with
 * SPS/PPS prefixed on sync samples.  Buffer owned by the demuxer, valid
 * until the next cc_demux_packet call or close. */
int  cc_demux_packet(cc_demux_t* d, size_t index, const uint8_t** pkt,
                     size_t* size, int64_t* pts, int32_t* keyframe);
void cc_demux_close(cc_demux_t* d);
/* Sample-exact stream-copy remux of the presentation span [start_s, end_s)
 * into a standalone MP4 (replaces the per-clip ffmpeg re-encode,
 * clip_extraction_stages.py:317-441; DESIGN.md "transcode").  Fails with
 * CC_ERR_UNSUPPORTED when the span's first sample is not a sync sample.
 * *out is malloc'd; free with cc_buffer_free. */
int  cc_demux_remux_clip(cc_demux_t* d, double start_s, double end_s,
                         uint8_t** out, size_t* out_size);
void cc_buffer_free(void* p);

/* ---- device memory (library-owned buffers; callers may also pass
 * torch-allocated device pointers to the kernels below) ---------------- */
int cc_malloc(void** dptr, size_t bytes);
int cc_free(void* dptr);
int cc_memcpy_h2d(void* dst, const void* src, size_t bytes, uint64_t stream);
int cc_memcpy_d2h(void* dst, const void* src, size_t bytes, uint64_t stream);
int cc_stream_sync(uint64_t stream);

/* ---- VCN hardware decode (rocDecode, runtime-probed) ----------------
 * Replaces NVDEC via PyNvVideoCodec CreateDecoder/Decode
 * (nvcodec_utils.py:199-313).  Fails with CC_ERR_NO_ROCDECODE when
 * librocdecode.so is absent (this image ships none; the ABI is the seam). */
typedef struct cc_decode cc_decode_t;
/* codec: 0 = h264, 1 = hevc (cc_video_info.codec values). */
int  cc_decode_session_create(int device, int32_t codec, cc_decode_t** out);
/* One AnnexB access unit (cc_demux_packet output) per call; pkt == NULL
 * flushes the stream (end-of-stream). */
int  cc_decode_submit(cc_decode_t* s, const uint8_t* pkt, size_t size, int64_t pts);
/* Mapped NV12 surfaces land as {y_ptr, uv_ptr, pitch} triples (HIP device
 * pointers into the decoder's surface pool).  Surfaces stay valid until
 * cc_decode_recycle; consume them (launch + synchronize) first. */
typedef struct cc_nv12_frame {
  void* y; void* uv; size_t pitch; int64_t pts; uint32_t width, height;
} cc_nv12_frame;
int  cc_decode_map_frames(cc_decode_t* s, cc_nv12_frame* out, size_t cap, size_t* n);
/* Return every surface handed out by map_frames to the decoder pool
 * (rocDecParserMarkFrameForReuse).  Call after the consuming kernels
 * complete. */
int  cc_decode_recycle(cc_decode_t* s);
void cc_decode_destroy(cc_decode_t* s);

/* ---- fused pixel kernels (hand-written HIP, gfx950) -----------------
 * Replace cvcuda.cvtcolor_into(YUV2RGB_NV12) (nvcodec_utils.py:178),
 * cvcuda.resize_into(LINEAR) (:189-194), cvcuda.reformat_into (:267),
 * cv2 INTER_CUBIC (decoder_utils.py:666-670) and the CLIP/torchvision
 * normalize chain (models/clip.py:48-62).  Batched NHWC u8 layouts. */

/* NV12 (Y plane + interleaved UV half-res plane, same pitch) ->
 * RGB888 NHWC at (out_h,out_w) via bilinear taps in converted-RGB space
 * (each tap converted BT.601 limited-range and rounded to u8 first —
 * bit-identical to convert-then-resize, fused to skip the intermediate). */
int cc_nv12_to_rgb_resize(const void* y, const void* uv, int n,
                          int src_h, int src_w, size_t pitch,
                          void* out_rgb, int out_h, int out_w,
                          uint64_t stream);
/* Plain NV12 -> RGB888 full resolution (the cvtcolor_into twin). */
int cc_nv12_to_rgb(const void* y, const void* uv, int n,
                   int h, int w, size_t pitch, void* out_rgb, uint64_t stream);
/* u8 NHWC bilinear resize (cvcuda LINEAR semantics). */
int cc_resize_bilinear_u8(const void* in, int n, int src_h, int src_w,
                          void* out, int dst_h, int dst_w, uint64_t stream);
/* u8 NHWC bicubic resize, A=-0.75 (cv2 INTER_CUBIC semantics). */
int cc_resize_bicubic_u8(const void* in, int n, int src_h, int src_w,
                         void* out, int dst_h, int dst_w, uint64_t stream);
/* (N,H,W,3) u8 -> (N,3,H,W) normalized ((x/255)-mean)/std.
 * out_dtype: 0 = f32, 1 = bf16. */
int cc_clip_preprocess(const void* in, int n, int h, int w,
                       const float mean[3], const float stdev[3],
                       void* out, int out_dtype, uint64_t stream);
/* Normalize + patch-extraction fused: u8 NHWC frames -> the ViT patch
 * GEMM's A layout [n*(h/patch)*(w/patch), kpad] bf16, col order
 * c*P*P + ky*P + kx (conv-weight flatten), zero-padded to kpad (64-
 * multiple).  Replaces the torch reshape/permute/pad chain after
 * cc_clip_preprocess. */
int cc_clip_preprocess_patches(const void* in, int n, int h, int w, int patch,
                               int kpad, const float mean[3],
                               const float stdev[3], void* out,
                               uint64_t stream);
/* Gather + duplicate-count broadcast of selected frames on device
 * (the decode loop's count broadcast, decoder_utils.py:447-453). */
int cc_gather_frames_u8(const void* frames, int n_in, size_t frame_bytes,
                        const int32_t* idx, const int32_t* counts, int n_idx,
                        int total_out, void* out, uint64_t stream);

/* ---- ViT MFMA GEMMs -------------------------------------------------
 * Replace the cuBLAS GEMMs under CLIPModel.get_image_features
 * (models/clip.py:71): patch-embed-as-GEMM, QKV/out projections, MLP.
 * C[M,N] = A[M,K] (bf16, row-major) x B[N,K]^T (bf16, row-major "weight
 * layout") + bias[N] (f32, optional NULL).  c_dtype: 0 = f32, 1 = bf16. */
int cc_gemm_bf16(const void* A, const void* B, void* C,
                 int64_t M, int64_t N, int64_t K,
                 const float* bias, int c_dtype, uint64_t stream);
/* Fused-epilogue form: act 0 = none, 1 = quick-gelu (x*sigmoid(1.702x),
 * the CLIP MLP activation); residual (bf16 [M,N], may be NULL) is added
 * after the activation — fuses the MLP/attention elementwise kernels into
 * the producing GEMM. */
int cc_gemm_bf16_ex(const void* A, const void* B, void* C,
                    int64_t M, int64_t N, int64_t K,
                    const float* bias, int c_dtype, int act,
                    const void* residual, uint64_t stream);

/* ---- fused short-sequence attention (ViT encoder; replaces torch sdpa
 * + the qkv permute copies).  qkv = the fused-QKV GEMM output
 * [n*seq, 3*hidden] bf16; out [n*seq, hidden] bf16; seq <= 64, head dim
 * 64. */
int cc_attn_small(const void* qkv, void* out, int64_t n_frames, int seq,
                  int heads, int hidden, float scale, uint64_t stream);

/* mid-sequence variant (64 < seq <= 288, head dim 64): ViT-L/14's
 * seq=257.  K/V^T LDS-resident per (frame, head) workgroup, Q in 64-row
 * tiles (replaces torch sdpa on that path). */
int cc_attn_mid(const void* qkv, void* out, int64_t n_frames, int seq,
                int heads, int hidden, float scale, uint64_t stream);

/* flash variant (seq > 288, head dim 64): K/V streamed in 64-row tiles
 * with online softmax — covers SigLIP-384/so400m-class towers. */
int cc_attn_flash(const void* qkv, void* out, int64_t n_frames, int seq,
                  int heads, int hidden, float scale, uint64_t stream);

/* ---- fused bf16 LayerNorm (replaces torch layer_norm in the ViT
 * forward; f32 stats/affine, H multiple of 256). */
int cc_layernorm_bf16(const void* x, const void* w, const void* b, void* y,
                      int64_t M, int64_t H, float eps, uint64_t stream);

/* ---- fused ViT embedding assembly: out[f*tokens + t] =
 * LayerNorm( (t==0 ? cls : tok[f*(tokens-1)+t-1]) + pos[t] ).
 * Replaces the HF CLIPVisionEmbeddings cat + position-embedding add +
 * pre_layrnorm chain (reference models/clip.py get_image_features
 * entry).  tok bf16 [n_frames*(tokens-1), H]; cls/pos/w/b f32; out bf16
 * [n_frames*tokens, H]; H multiple of 256. */
int cc_embed_assemble_ln(const void* tok, const void* cls, const void* pos,
                         const void* w, const void* b, void* y,
                         int64_t n_frames, int64_t tokens, int64_t H,
                         float eps, uint64_t stream);

/* ---- semantic dedup -------------------------------------------------
 * Strict-upper-triangular max-cosine scan (SemDedupActor.dedup,
 * pipelines/video/dedup/dedup_actor.py:315-460): for each row j of the
 * L2-normalized f32 matrix e[m][d] (scan order), the max cosine to any
 * earlier row and its argmax.  Row 0 -> (0.0, 0).  d % 64 == 0. */
int cc_pairwise_max_earlier(const void* e_f32, int64_t m, int64_t d,
                            void* maxv_f32, void* argi_i32, uint64_t stream);

/* ---- kernel timing (bench.py roofline evidence) ---------------------
 * When enabled, every cc_* kernel launch is bracketed with hipEvents on
 * its launch stream; totals are accumulated per kernel name. */
int cc_timing_enable(int enable);
int cc_timing_reset(void);
/* Fetch totals for `kernel` (e.g. "gemm_bf16"): total device ms + count. */
int cc_timing_report(const char* kernel, double* total_ms, int64_t* count);

#ifdef __cplusplus
}
#endif
#endif /* CC_HOTPATH_H */
