// VCN hardware-decode seam (rocDecode, runtime-probed).
//
// Replaces NVDEC via PyNvVideoCodec (/root/reference/cosmos_curate/
// pipelines/video/utils/nvcodec_utils.py:199-313).  This ROCm image ships
// no librocdecode, so the session functions probe for it with dlopen at
// runtime and fail loudly with CC_ERR_NO_ROCDECODE when absent — the
// kernel path downstream of decode (NV12 surfaces in HBM) is exercised
// with raw-frame fixtures instead (SURVEY.md §7 hard part (d): pluggable
// codec backend).  When a box with librocdecode.so appears, only this file
// grows a real session implementation; the ABI and everything downstream
// stay fixed.

#include <dlfcn.h>

#include "cc_common.hpp"

namespace {

void* rocdecode_handle() {
  static void* h = [] {
    void* p = dlopen("librocdecode.so", RTLD_NOW | RTLD_LOCAL);
    if (!p) p = dlopen("librocdecode.so.0", RTLD_NOW | RTLD_LOCAL);
    return p;
  }();
  return h;
}

}  // namespace

struct cc_decode {
  int device;
  int32_t codec;
};

extern "C" {

int cc_rocdecode_available(void) {
  if (!rocdecode_handle())
    return cc::set_error(CC_ERR_NO_ROCDECODE,
                         "librocdecode.so not found (VCN decode unavailable)");
  return CC_OK;
}

int cc_decode_session_create(int device, int32_t codec, cc_decode** out) {
  if (!out) return cc::set_error(CC_ERR_INVALID, "null out");
  int rc = cc_rocdecode_available();
  if (rc != CC_OK) return rc;  // loud failure; no CPU fallback
  // Real rocDecode session wiring lands when a librocdecode-equipped box
  // exists to validate against; until then reaching this line is
  // unreachable in practice.
  return cc::set_error(CC_ERR_UNSUPPORTED,
                       "rocDecode session wiring not yet implemented");
}

int cc_decode_submit(cc_decode* s, const uint8_t* pkt, size_t size, int64_t pts) {
  (void)s; (void)pkt; (void)size; (void)pts;
  return cc::set_error(CC_ERR_UNSUPPORTED, "no decode session");
}

int cc_decode_map_frames(cc_decode* s, cc_nv12_frame* out, size_t cap, size_t* n) {
  (void)s; (void)out; (void)cap; (void)n;
  return cc::set_error(CC_ERR_UNSUPPORTED, "no decode session");
}

void cc_decode_destroy(cc_decode* s) { delete s; }

}  // extern "C"
