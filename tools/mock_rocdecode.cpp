// Mock librocdecode.so — exercises csrc/cc_decode.cpp's full session
// wiring without VCN hardware (tests/test_decode_session.py builds this
// and injects it via LD_LIBRARY_PATH in a subprocess).
//
// Simulates a 48x32 8-bit AVC stream against the PUBLIC rocDecode API
// (rocprofiler-sdk/rocdecode/details/*.h): the first data packet
// triggers the sequence callback, every packet then triggers one decode
// + one display callback; GetVideoFrame hands out distinct fake device
// pointers.  Introspection counters are exported as mock_* symbols.
//
//   g++ -shared -fPIC -D__HIP_PLATFORM_AMD__ \
//       -I/opt/rocm/include tools/mock_rocdecode.cpp -o librocdecode.so

#include <cstring>

#define __HIP_PLATFORM_AMD__ 1
#include "rocprofiler-sdk/rocdecode/details/rocdecode.h"
#include "rocprofiler-sdk/rocdecode/details/rocparser.h"

namespace {

struct MockParser {
  RocdecParserParams params;
  int packets = 0;
  int pic_counter = 0;
  int dpb = 0;
};

struct MockDecoder {
  RocDecoderCreateInfo info;
};

int g_reused = 0;       // MarkFrameForReuse calls
int g_decoded = 0;      // DecodeFrame calls
int g_destroyed = 0;    // parser+decoder destroys
int g_eos_seen = 0;

}  // namespace

extern "C" {

// test introspection
int mock_reused(void) { return g_reused; }
int mock_decoded(void) { return g_decoded; }
int mock_destroyed(void) { return g_destroyed; }
int mock_eos_seen(void) { return g_eos_seen; }

rocDecStatus rocDecCreateVideoParser(RocdecVideoParser* h,
                                     RocdecParserParams* p) {
  if (!h || !p || !p->pfn_sequence_callback || !p->pfn_decode_picture ||
      !p->pfn_display_picture)
    return ROCDEC_INVALID_PARAMETER;
  auto* m = new MockParser();
  m->params = *p;
  m->dpb = (int)p->max_num_decode_surfaces;
  *h = m;
  return ROCDEC_SUCCESS;
}

rocDecStatus rocDecParseVideoData(RocdecVideoParser h,
                                  RocdecSourceDataPacket* pkt) {
  auto* m = (MockParser*)h;
  if (!m || !pkt) return ROCDEC_INVALID_PARAMETER;
  if (pkt->flags & ROCDEC_PKT_ENDOFSTREAM) {
    g_eos_seen = 1;
    if (pkt->flags & ROCDEC_PKT_NOTIFY_EOS)
      m->params.pfn_display_picture(m->params.user_data, nullptr);
    return ROCDEC_SUCCESS;
  }
  if (!pkt->payload || !pkt->payload_size) return ROCDEC_INVALID_PARAMETER;
  if (m->packets == 0) {
    RocdecVideoFormat fmt = {};
    fmt.codec = m->params.codec_type;
    fmt.progressive_sequence = 1;
    fmt.min_num_decode_surfaces = 4;
    fmt.coded_width = 48;
    fmt.coded_height = 32;
    fmt.display_area.left = 0;
    fmt.display_area.top = 0;
    fmt.display_area.right = 48;
    fmt.display_area.bottom = 30;  // display crop (like 1088 -> 1080)
    fmt.chroma_format = rocDecVideoChromaFormat_420;
    int r = m->params.pfn_sequence_callback(m->params.user_data, &fmt);
    if (r == 0) return ROCDEC_RUNTIME_ERROR;
    if (r > 1) m->dpb = r;
  }
  m->packets++;
  RocdecPicParams pic = {};
  pic.pic_width = 48;
  pic.pic_height = 32;
  pic.curr_pic_idx = m->pic_counter % m->dpb;
  if (m->params.pfn_decode_picture(m->params.user_data, &pic) == 0)
    return ROCDEC_RUNTIME_ERROR;
  RocdecParserDispInfo disp = {};
  disp.picture_index = m->pic_counter % m->dpb;
  disp.progressive_frame = 1;
  disp.pts = (pkt->flags & ROCDEC_PKT_TIMESTAMP) ? pkt->pts : 0;
  m->pic_counter++;
  if (m->params.pfn_display_picture(m->params.user_data, &disp) == 0)
    return ROCDEC_RUNTIME_ERROR;
  return ROCDEC_SUCCESS;
}

rocDecStatus rocDecDestroyVideoParser(RocdecVideoParser h) {
  delete (MockParser*)h;
  g_destroyed++;
  return ROCDEC_SUCCESS;
}

rocDecStatus rocDecParserMarkFrameForReuse(RocdecVideoParser h, int pic_idx) {
  if (!h || pic_idx < 0) return ROCDEC_INVALID_PARAMETER;
  g_reused++;
  return ROCDEC_SUCCESS;
}

rocDecStatus rocDecCreateDecoder(rocDecDecoderHandle* h,
                                 RocDecoderCreateInfo* info) {
  if (!h || !info || info->width == 0) return ROCDEC_INVALID_PARAMETER;
  if (info->output_format != rocDecVideoSurfaceFormat_NV12)
    return ROCDEC_NOT_SUPPORTED;
  auto* d = new MockDecoder();
  d->info = *info;
  *h = d;
  return ROCDEC_SUCCESS;
}

rocDecStatus rocDecDestroyDecoder(rocDecDecoderHandle h) {
  delete (MockDecoder*)h;
  g_destroyed++;
  return ROCDEC_SUCCESS;
}

rocDecStatus rocDecDecodeFrame(rocDecDecoderHandle h, RocdecPicParams* pic) {
  if (!h || !pic) return ROCDEC_INVALID_PARAMETER;
  g_decoded++;
  return ROCDEC_SUCCESS;
}

rocDecStatus rocDecGetVideoFrame(rocDecDecoderHandle h, int pic_idx,
                                 void* planes[3], uint32_t* pitch,
                                 RocdecProcParams* proc) {
  if (!h || !planes || !pitch || !proc || pic_idx < 0)
    return ROCDEC_INVALID_PARAMETER;
  planes[0] = (void*)(uintptr_t)(0x100000 + pic_idx * 0x10000);
  planes[1] = (void*)(uintptr_t)(0x100000 + pic_idx * 0x10000 + 0x8000);
  planes[2] = nullptr;
  pitch[0] = 64;
  pitch[1] = 64;
  pitch[2] = 0;
  return ROCDEC_SUCCESS;
}

rocDecStatus rocDecReconfigureDecoder(rocDecDecoderHandle h,
                                      RocdecReconfigureDecoderInfo* rc) {
  if (!h || !rc) return ROCDEC_INVALID_PARAMETER;
  return ROCDEC_SUCCESS;
}

const char* rocDecGetErrorName(rocDecStatus s) {
  switch (s) {
    case ROCDEC_SUCCESS: return "ROCDEC_SUCCESS";
    case ROCDEC_INVALID_PARAMETER: return "ROCDEC_INVALID_PARAMETER";
    case ROCDEC_RUNTIME_ERROR: return "ROCDEC_RUNTIME_ERROR";
    case ROCDEC_NOT_SUPPORTED: return "ROCDEC_NOT_SUPPORTED";
    default: return "ROCDEC_ERROR";
  }
}

}  // extern "C"
