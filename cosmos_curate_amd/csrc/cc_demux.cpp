// Host MP4 (ISO BMFF) demuxer — the PTS + AnnexB packet source of the
// rebuilt hot path.
//
// Replaces PyNvDemuxer (/root/reference/cosmos_curate/pipelines/video/utils/
// nvcodec_utils.py:224) and implements the PTS contract of
// get_video_timestamps (decoder_utils.py:230-278): per-sample presentation
// timestamps from stts/ctts, shifted by the first non-empty elst edit
// (libavformat mov semantics), converted to float32 seconds and sorted.
//
// Pure host C++ — runs without a GPU.  Parity: tests/test_demux_abi.py
// checks this against oracle/mp4_demux.py on the committed fixtures.

#include <algorithm>
#include <cstdlib>
#include <cstring>
#include <vector>

#include "cc_common.hpp"

namespace {

struct Box {
  char type[5];
  size_t body_start, body_end;
};

// iterate boxes in data[start,end)
struct BoxIter {
  const uint8_t* data;
  size_t pos, end;
  bool next(Box* out) {
    while (pos + 8 <= end) {
      uint64_t size = (uint64_t)((data[pos] << 24) | (data[pos + 1] << 16) |
                                 (data[pos + 2] << 8) | data[pos + 3]);
      size_t hdr = 8;
      if (size == 1) {
        if (pos + 16 > end) return false;
        size = 0;
        for (int i = 0; i < 8; i++) size = (size << 8) | data[pos + 8 + i];
        hdr = 16;
      } else if (size == 0) {
        size = end - pos;
      }
      if (size < hdr || pos + size > end) return false;
      memcpy(out->type, data + pos + 4, 4);
      out->type[4] = 0;
      out->body_start = pos + hdr;
      out->body_end = pos + size;
      pos += size;
      return true;
    }
    return false;
  }
};

inline uint32_t rd32(const uint8_t* p) {
  return ((uint32_t)p[0] << 24) | (p[1] << 16) | (p[2] << 8) | p[3];
}
inline int32_t rd32s(const uint8_t* p) { return (int32_t)rd32(p); }

// robustness caps for corrupt containers (fuzz contract: clean error or
// bounded parse, never a hang / OOB / memory blowup)
constexpr size_t CC_MAX_SAMPLES = 1u << 24;

struct Box;  // fwd
template <typename BoxT>
inline uint32_t clamp_entries(uint32_t n, const BoxT& bx, size_t hdr,
                              size_t entry) {
  size_t body = bx.body_end > bx.body_start + hdr
                    ? bx.body_end - bx.body_start - hdr
                    : 0;
  size_t fit = body / entry;
  return n < fit ? n : (uint32_t)fit;
}
inline uint16_t rd16(const uint8_t* p) { return (uint16_t)((p[0] << 8) | p[1]); }
inline uint64_t rd64(const uint8_t* p) {
  return ((uint64_t)rd32(p) << 32) | rd32(p + 4);
}

}  // namespace

struct cc_demux {
  std::vector<uint8_t> data;
  // sample tables (first video track)
  uint32_t timescale = 0;
  int32_t codec = -1;  // 0 h264, 1 hevc
  uint32_t width = 0, height = 0;
  std::vector<uint32_t> sizes;
  std::vector<uint64_t> offsets;
  std::vector<int64_t> dts;
  std::vector<int32_t> cts;
  std::vector<uint32_t> sync;  // 1-based sample numbers; empty = all sync
  int64_t elst_media_time = 0;
  std::vector<uint8_t> avcc;  // avcC / hvcC record
  int nal_len_size = 4;
  std::vector<uint8_t> ps_prefix;  // SPS/PPS AnnexB prefix
  std::vector<uint8_t> pkt_buf;    // scratch for cc_demux_packet
};

static bool parse_stbl(const uint8_t* d, size_t b, size_t e, cc_demux* t) {
  bool bad = false;  // corrupt-table flag: normalize, then reject
  std::vector<std::pair<uint32_t, uint32_t>> stsc;  // first_chunk, samples_per_chunk
  std::vector<uint64_t> chunk_offsets;
  BoxIter it{d, b, e};
  Box bx;
  while (it.next(&bx)) {
    const uint8_t* p = d + bx.body_start;
    if (!strcmp(bx.type, "stsd")) {
      uint32_t n = rd32(p + 4);
      size_t pos = bx.body_start + 8;
      for (uint32_t i = 0; i < n && pos + 8 <= bx.body_end; i++) {
        uint32_t esize = rd32(d + pos);
        // corrupt entry sizes must neither escape the box nor stall the
        // loop (esize 0) nor let the nested iterator read past the box
        if (esize < 16 || esize > bx.body_end - pos) break;
        char fmt[5] = {0};
        memcpy(fmt, d + pos + 4, 4);
        if (!strcmp(fmt, "avc1") || !strcmp(fmt, "avc3")) t->codec = 0;
        if (!strcmp(fmt, "hvc1") || !strcmp(fmt, "hev1")) t->codec = 1;
        if (t->codec >= 0 && esize >= 86) {
          t->width = rd16(d + pos + 32);
          t->height = rd16(d + pos + 34);
          BoxIter it2{d, pos + 86, pos + esize};
          Box b2;
          while (it2.next(&b2)) {
            if (!strcmp(b2.type, "avcC") || !strcmp(b2.type, "hvcC"))
              t->avcc.assign(d + b2.body_start, d + b2.body_end);
          }
        }
        pos += esize;
      }
    } else if (!strcmp(bx.type, "stts")) {
      // corrupt-input bounds (fuzz-found hang: a flipped count byte made
      // the run expansion loop effectively unbounded): entry lists are
      // clamped to what fits in the box, expansions to CC_MAX_SAMPLES.
      uint32_t n = clamp_entries(rd32(p + 4), bx, 8, 8);
      int64_t tcur = 0;
      size_t pos = bx.body_start + 8;
      for (uint32_t i = 0; i < n; i++, pos += 8) {
        uint32_t cnt = rd32(d + pos), delta = rd32(d + pos + 4);
        for (uint32_t j = 0; j < cnt; j++) {
          if (t->dts.size() >= CC_MAX_SAMPLES) { bad = true; break; }
          t->dts.push_back(tcur);
          tcur += delta;
        }
        if (bad) break;
      }
    } else if (!strcmp(bx.type, "ctts")) {
      uint32_t n = clamp_entries(rd32(p + 4), bx, 8, 8);
      size_t pos = bx.body_start + 8;
      for (uint32_t i = 0; i < n; i++, pos += 8) {
        uint32_t cnt = rd32(d + pos);
        int32_t off = rd32s(d + pos + 4);
        for (uint32_t j = 0; j < cnt; j++) {
          if (t->cts.size() >= CC_MAX_SAMPLES) { bad = true; break; }
          t->cts.push_back(off);
        }
        if (bad) break;
      }
    } else if (!strcmp(bx.type, "stss")) {
      uint32_t n = clamp_entries(rd32(p + 4), bx, 8, 4);
      for (uint32_t i = 0; i < n; i++) t->sync.push_back(rd32(p + 8 + 4 * i));
    } else if (!strcmp(bx.type, "stsz")) {
      uint32_t ss = rd32(p + 4), n = rd32(p + 8);
      if (ss) {
        if (n > CC_MAX_SAMPLES) { bad = true; n = 0; }
        t->sizes.assign(n, ss);
      } else {
        n = clamp_entries(n, bx, 12, 4);
        for (uint32_t i = 0; i < n; i++) t->sizes.push_back(rd32(p + 12 + 4 * i));
      }
    } else if (!strcmp(bx.type, "stsc")) {
      uint32_t n = clamp_entries(rd32(p + 4), bx, 8, 12);
      for (uint32_t i = 0; i < n; i++)
        stsc.push_back({rd32(p + 8 + 12 * i), rd32(p + 12 + 12 * i)});
    } else if (!strcmp(bx.type, "stco") || !strcmp(bx.type, "co64")) {
      bool w = !strcmp(bx.type, "co64");
      uint32_t n = clamp_entries(rd32(p + 4), bx, 8, w ? 8 : 4);
      for (uint32_t i = 0; i < n; i++)
        chunk_offsets.push_back(w ? rd64(p + 8 + 8 * i) : rd32(p + 8 + 4 * i));
    }
  }
  if (t->cts.empty()) t->cts.assign(t->dts.size(), 0);
  // corrupt containers can leave the tables inconsistent: normalize so
  // every per-sample array shares one bounded length (no OOB indexing
  // anywhere downstream)
  t->cts.resize(t->dts.size(), 0);
  // expand chunk map to per-sample offsets (clamped: corrupt stsc
  // first_chunk fields must not produce unbounded runs)
  if (!chunk_offsets.empty() && !stsc.empty() && !t->sizes.empty()) {
    std::vector<uint32_t> per_chunk;
    for (size_t i = 0; i < stsc.size(); i++) {
      uint32_t last = (i + 1 < stsc.size()) ? stsc[i + 1].first - 1
                                            : (uint32_t)chunk_offsets.size();
      if (last > chunk_offsets.size()) last = (uint32_t)chunk_offsets.size();
      for (uint32_t c = stsc[i].first; c != 0 && c <= last; c++)
        per_chunk.push_back(stsc[i].second);
    }
    size_t si = 0;
    for (size_t ci = 0; ci < chunk_offsets.size(); ci++) {
      uint64_t pos = chunk_offsets[ci];
      uint32_t cnt = ci < per_chunk.size() ? per_chunk[ci] : 0;
      for (uint32_t j = 0; j < cnt && si < t->sizes.size(); j++, si++) {
        t->offsets.push_back(pos);
        pos += t->sizes[si];
      }
    }
  }
  // one consistent sample count across all per-sample tables
  size_t ns = t->dts.size();
  ns = std::min(ns, t->sizes.size());
  ns = std::min(ns, t->offsets.size());
  t->dts.resize(ns);
  t->cts.resize(ns, 0);
  t->sizes.resize(ns);
  t->offsets.resize(ns);
  return !bad;
}

static bool parse_trak(const uint8_t* d, size_t b, size_t e, cc_demux* t) {
  bool is_video = false;
  BoxIter it{d, b, e};
  Box bx;
  while (it.next(&bx)) {
    const uint8_t* p = d + bx.body_start;
    if (!strcmp(bx.type, "edts")) {
      BoxIter it2{d, bx.body_start, bx.body_end};
      Box b2;
      while (it2.next(&b2)) {
        if (!strcmp(b2.type, "elst") && b2.body_end >= b2.body_start + 8) {
          const uint8_t* q = d + b2.body_start;
          uint8_t ver = q[0];
          uint32_t n = rd32(q + 4);
          size_t pos = b2.body_start + 8;
          const size_t esz = (ver == 1) ? 20 : 12;
          for (uint32_t i = 0; i < n && pos + esz <= b2.body_end; i++) {
            int64_t media_time;
            if (ver == 1) {
              media_time = (int64_t)rd64(d + pos + 8);
              pos += 20;
            } else {
              media_time = rd32s(d + pos + 4);
              pos += 12;
            }
            if (media_time != -1) {
              t->elst_media_time = media_time;
              break;
            }
          }
        }
      }
    } else if (!strcmp(bx.type, "mdia")) {
      BoxIter it2{d, bx.body_start, bx.body_end};
      Box b2;
      while (it2.next(&b2)) {
        const uint8_t* q = d + b2.body_start;
        if (!strcmp(b2.type, "mdhd") && b2.body_end >= b2.body_start + 24) {
          uint8_t ver = q[0];
          t->timescale = rd32(q + (ver == 1 ? 20 : 12));
        } else if (!strcmp(b2.type, "hdlr") &&
                   b2.body_end >= b2.body_start + 12) {
          if (!memcmp(q + 8, "vide", 4)) is_video = true;
        } else if (!strcmp(b2.type, "minf")) {
          BoxIter it3{d, b2.body_start, b2.body_end};
          Box b3;
          while (it3.next(&b3))
            if (!strcmp(b3.type, "stbl") &&
                !parse_stbl(d, b3.body_start, b3.body_end, t))
              return false;  // corrupt sample tables: reject the track
        }
      }
    }
  }
  return is_video && t->timescale > 0;
}

static void build_ps_prefix(cc_demux* t) {
  // AnnexB parameter-set prefix from avcC (h264) or hvcC (hevc)
  static const uint8_t sc[4] = {0, 0, 0, 1};
  const uint8_t* a = t->avcc.data();
  size_t pos = 0;
  auto append_sets = [&](int count) {
    for (int i = 0; i < count && pos + 2 <= t->avcc.size(); i++) {
      uint16_t ln = rd16(a + pos);
      pos += 2;
      if (pos + ln > t->avcc.size()) return;
      t->ps_prefix.insert(t->ps_prefix.end(), sc, sc + 4);
      t->ps_prefix.insert(t->ps_prefix.end(), a + pos, a + pos + ln);
      pos += ln;
    }
  };
  if (t->codec == 0 && t->avcc.size() >= 7) {
    t->nal_len_size = (a[4] & 0x03) + 1;
    int num_sps = a[5] & 0x1F;
    pos = 6;  // skip the SPS-count byte
    append_sets(num_sps);
    if (pos < t->avcc.size()) {
      int num_pps = a[pos++];
      append_sets(num_pps);
    }
  } else if (t->codec == 1 && t->avcc.size() >= 23) {
    // HEVCDecoderConfigurationRecord (ISO 14496-15 §8.3.3.1): 22-byte
    // header (byte 21 low bits = lengthSizeMinusOne), numOfArrays at 22,
    // then arrays of (1-byte type, 2-byte count, count x (len, nalu)) —
    // VPS/SPS/PPS all go into the sync-sample prefix.
    t->nal_len_size = (a[21] & 0x03) + 1;
    int num_arrays = a[22];
    pos = 23;
    for (int ar = 0; ar < num_arrays && pos + 3 <= t->avcc.size(); ar++) {
      int cnt = rd16(a + pos + 1);
      pos += 3;
      append_sets(cnt);
    }
  }
}

extern "C" {

const char* cc_last_error(void) { return cc::g_last_error.c_str(); }

int cc_demux_open(const uint8_t* data, size_t size, cc_demux** out) {
  if (!data || !size || !out) return cc::set_error(CC_ERR_INVALID, "null arg");
  auto* t = new cc_demux();
  t->data.assign(data, data + size);
  const uint8_t* d = t->data.data();
  BoxIter it{d, 0, size};
  Box bx;
  bool found = false;
  while (it.next(&bx) && !found) {
    if (!strcmp(bx.type, "moov")) {
      BoxIter it2{d, bx.body_start, bx.body_end};
      Box b2;
      while (it2.next(&b2)) {
        if (!strcmp(b2.type, "trak")) {
          cc_demux probe;
          if (parse_trak(d, b2.body_start, b2.body_end, &probe) &&
              probe.codec >= 0) {
            // keep tables, move into t
            t->timescale = probe.timescale;
            t->codec = probe.codec;
            t->width = probe.width;
            t->height = probe.height;
            t->sizes = std::move(probe.sizes);
            t->offsets = std::move(probe.offsets);
            t->dts = std::move(probe.dts);
            t->cts = std::move(probe.cts);
            t->sync = std::move(probe.sync);
            t->elst_media_time = probe.elst_media_time;
            t->avcc = std::move(probe.avcc);
            found = true;
            break;
          }
        }
      }
    }
  }
  if (!found || t->dts.empty()) {
    delete t;
    return cc::set_error(CC_ERR_PARSE, "no decodable video track");
  }
  build_ps_prefix(t);
  *out = t;
  return CC_OK;
}

int cc_demux_probe(const cc_demux* d, cc_video_info* info) {
  if (!d || !info) return cc::set_error(CC_ERR_INVALID, "null arg");
  info->width = d->width;
  info->height = d->height;
  info->timescale = d->timescale;
  info->num_samples = (uint32_t)d->dts.size();
  info->num_sync_samples = (uint32_t)d->sync.size();
  info->codec = d->codec;
  int64_t span = d->dts.empty() ? 0 : d->dts.back() - d->dts.front();
  // media duration: last dts + last delta ~= samples * avg delta
  double dur = d->dts.size() > 1
                   ? (double)span / (d->dts.size() - 1) * d->dts.size() / d->timescale
                   : 0.0;
  info->duration_s = dur;
  info->avg_fps = dur > 0 ? d->dts.size() / dur : 0.0;
  return CC_OK;
}

int cc_demux_timestamps(const cc_demux* d, float* out, size_t cap, size_t* n) {
  if (!d || !n) return cc::set_error(CC_ERR_INVALID, "null arg");
  *n = d->dts.size();
  if (!out) return CC_OK;  // size query
  if (cap < d->dts.size()) return cc::set_error(CC_ERR_INVALID, "cap too small");
  std::vector<float> ts(d->dts.size());
  for (size_t i = 0; i < d->dts.size(); i++) {
    // float32 of (pts - elst_shift) / timescale, exactly as
    // decoder_utils.py:275 computes float(pts) * time_base in f64 then
    // narrows to f32 on array construction.
    double sec = (double)(d->dts[i] + d->cts[i] - d->elst_media_time) /
                 (double)d->timescale;
    ts[i] = (float)sec;
  }
  std::sort(ts.begin(), ts.end());
  memcpy(out, ts.data(), ts.size() * sizeof(float));
  return CC_OK;
}

int cc_demux_packet(cc_demux* d, size_t index, const uint8_t** pkt, size_t* size,
                    int64_t* pts, int32_t* keyframe) {
  if (!d || !pkt || !size) return cc::set_error(CC_ERR_INVALID, "null arg");
  if (index >= d->sizes.size() || index >= d->offsets.size())
    return cc::set_error(CC_ERR_INVALID, "sample index out of range");
  size_t ssize = d->sizes[index];
  if (d->offsets[index] > d->data.size() ||
      ssize > d->data.size() - d->offsets[index])
    return cc::set_error(CC_ERR_PARSE, "sample range outside file");
  const uint8_t* sample = d->data.data() + d->offsets[index];
  bool is_sync = d->sync.empty() ||
                 std::binary_search(d->sync.begin(), d->sync.end(), (uint32_t)(index + 1));
  d->pkt_buf.clear();
  if (is_sync)
    d->pkt_buf.insert(d->pkt_buf.end(), d->ps_prefix.begin(), d->ps_prefix.end());
  static const uint8_t sc[4] = {0, 0, 0, 1};
  size_t p = 0;
  while (p + d->nal_len_size <= ssize) {
    uint64_t ln = 0;
    for (int i = 0; i < d->nal_len_size; i++) ln = (ln << 8) | sample[p + i];
    p += d->nal_len_size;
    size_t take = std::min<uint64_t>(ln, ssize - p);
    d->pkt_buf.insert(d->pkt_buf.end(), sc, sc + 4);
    d->pkt_buf.insert(d->pkt_buf.end(), sample + p, sample + p + take);
    p += take;
  }
  *pkt = d->pkt_buf.data();
  *size = d->pkt_buf.size();
  if (pts) *pts = d->dts[index] + d->cts[index] - d->elst_media_time;
  if (keyframe) *keyframe = is_sync ? 1 : 0;
  return CC_OK;
}

void cc_demux_close(cc_demux* d) { delete d; }

}  // extern "C"

// ---- stream-copy clip remux ------------------------------------------
// Replaces the per-clip ffmpeg re-encode of ClipTranscodingStage
// (clip_extraction_stages.py:317-441) with a sample-exact remux: the
// samples whose presentation time falls in [start_s, end_s) are copied
// bit-for-bit into a fresh minimal MP4 (ftyp/moov/mdat, one chunk),
// carrying the source avcC, rebased stts/ctts and remapped stss.
// Requires the span to start on a sync sample (no re-encode fallback —
// the error is recorded per clip upstream).

namespace {

struct ByteVec {
  std::vector<uint8_t> v;
  void u8(uint8_t x) { v.push_back(x); }
  void u16(uint16_t x) { v.push_back(x >> 8); v.push_back(x & 0xff); }
  void u32(uint32_t x) {
    v.push_back(x >> 24); v.push_back((x >> 16) & 0xff);
    v.push_back((x >> 8) & 0xff); v.push_back(x & 0xff);
  }
  void bytes(const uint8_t* p, size_t n) { v.insert(v.end(), p, p + n); }
  void zeros(size_t n) { v.insert(v.end(), n, 0); }
};

// box with size patched at close
struct BoxW {
  ByteVec& b;
  size_t at;
  BoxW(ByteVec& bv, const char* typ) : b(bv), at(bv.v.size()) {
    b.u32(0);
    b.bytes((const uint8_t*)typ, 4);
  }
  void close() {
    uint32_t sz = (uint32_t)(b.v.size() - at);
    b.v[at] = sz >> 24; b.v[at + 1] = (sz >> 16) & 0xff;
    b.v[at + 2] = (sz >> 8) & 0xff; b.v[at + 3] = sz & 0xff;
  }
};

}  // namespace

extern "C" {

void cc_buffer_free(void* p) { free(p); }

int cc_demux_remux_clip(cc_demux* d, double start_s, double end_s,
                        uint8_t** out, size_t* out_size) {
  if (!d || !out || !out_size) return cc::set_error(CC_ERR_INVALID, "null arg");
  if ((d->codec != 0 && d->codec != 1) || d->avcc.empty())
    return cc::set_error(CC_ERR_UNSUPPORTED,
                         "remux needs an avcC/hvcC h264 or hevc track");
  const size_t n = d->dts.size();
  if (d->offsets.size() != n || d->sizes.size() != n)
    return cc::set_error(CC_ERR_PARSE, "incomplete sample tables");
  // select samples by presentation seconds (decode order scan)
  int64_t lo = -1, hi = -1;  // [lo, hi] inclusive decode-order range
  for (size_t i = 0; i < n; i++) {
    double pts = (double)(d->dts[i] + d->cts[i] - d->elst_media_time) / d->timescale;
    if (pts >= start_s - 1e-9 && pts < end_s - 1e-9) {
      if (lo < 0) lo = (int64_t)i;
      hi = (int64_t)i;
    }
  }
  if (lo < 0) return cc::set_error(CC_ERR_INVALID, "empty span");
  bool lo_sync = d->sync.empty() ||
                 std::binary_search(d->sync.begin(), d->sync.end(), (uint32_t)(lo + 1));
  if (!lo_sync)
    return cc::set_error(CC_ERR_UNSUPPORTED,
                         "span start (sample %lld) is not a sync sample; "
                         "stream-copy needs a keyframe-aligned span",
                         (long long)lo);
  const size_t m = (size_t)(hi - lo + 1);

  // sample data (length-prefixed, copied verbatim) + new tables
  ByteVec mdat_payload;
  std::vector<uint32_t> sizes(m);
  uint64_t total_dur = 0;
  std::vector<std::pair<uint32_t, uint32_t>> stts;  // (count, delta) runs
  for (size_t i = 0; i < m; i++) {
    size_t si = lo + i;
    if (d->offsets[si] > d->data.size() ||
        d->sizes[si] > d->data.size() - d->offsets[si])
      return cc::set_error(CC_ERR_PARSE, "sample range outside file");
    sizes[i] = d->sizes[si];
    mdat_payload.bytes(d->data.data() + d->offsets[si], d->sizes[si]);
    uint32_t delta;
    if (si + 1 < n)
      delta = (uint32_t)(d->dts[si + 1] - d->dts[si]);
    else if (si > 0)
      delta = (uint32_t)(d->dts[si] - d->dts[si - 1]);
    else
      delta = d->timescale / 30;
    total_dur += delta;
    if (!stts.empty() && stts.back().second == delta)
      stts.back().first++;
    else
      stts.push_back({1, delta});
  }

  auto build_moov = [&](uint32_t chunk_off) {
    ByteVec b;
    BoxW moov(b, "moov");
    {
      BoxW mvhd(b, "mvhd");
      b.u32(0);  // version+flags
      b.u32(0); b.u32(0);
      b.u32(d->timescale); b.u32((uint32_t)total_dur);
      b.u32(0x00010000); b.u16(0x0100); b.u16(0); b.u32(0); b.u32(0);
      const int32_t mat[9] = {0x10000, 0, 0, 0, 0x10000, 0, 0, 0, 0x40000000};
      for (int32_t x : mat) b.u32((uint32_t)x);
      b.zeros(24);
      b.u32(2);  // next track id
      mvhd.close();
    }
    {
      BoxW trak(b, "trak");
      {
        BoxW tkhd(b, "tkhd");
        b.u32(7);  // version 0, flags enabled
        b.u32(0); b.u32(0); b.u32(1); b.u32(0); b.u32((uint32_t)total_dur);
        b.zeros(16);
        const int32_t mat[9] = {0x10000, 0, 0, 0, 0x10000, 0, 0, 0, 0x40000000};
        for (int32_t x : mat) b.u32((uint32_t)x);
        b.u32((uint32_t)d->width << 16); b.u32((uint32_t)d->height << 16);
        tkhd.close();
      }
      {
        BoxW mdia(b, "mdia");
        {
          BoxW mdhd(b, "mdhd");
          b.u32(0); b.u32(0); b.u32(0);
          b.u32(d->timescale); b.u32((uint32_t)total_dur);
          b.u16(0x55c4); b.u16(0);
          mdhd.close();
        }
        {
          BoxW hdlr(b, "hdlr");
          b.u32(0); b.u32(0);
          b.bytes((const uint8_t*)"vide", 4);
          b.zeros(12);
          b.bytes((const uint8_t*)"v\0", 2);
          hdlr.close();
        }
        {
          BoxW minf(b, "minf");
          {
            BoxW vmhd(b, "vmhd");
            b.u32(1); b.zeros(8);
            vmhd.close();
          }
          {
            BoxW dinf(b, "dinf");
            BoxW dref(b, "dref");
            b.u32(0); b.u32(1);
            BoxW url(b, "url ");
            b.u32(1);
            url.close();
            dref.close();
            dinf.close();
          }
          {
            BoxW stbl(b, "stbl");
            {
              BoxW stsd(b, "stsd");
              b.u32(0); b.u32(1);
              // entry + config box types follow the source codec
              BoxW avc1(b, d->codec == 1 ? "hvc1" : "avc1");
              b.zeros(6); b.u16(1);
              b.zeros(16);
              b.u16((uint16_t)d->width); b.u16((uint16_t)d->height);
              b.u32(0x00480000); b.u32(0x00480000); b.u32(0);
              b.u16(1); b.zeros(32);
              b.u16(0x18); b.u16(0xffff);
              BoxW avcC(b, d->codec == 1 ? "hvcC" : "avcC");
              b.bytes(d->avcc.data(), d->avcc.size());
              avcC.close();
              avc1.close();
              stsd.close();
            }
            {
              BoxW stts_b(b, "stts");
              b.u32(0); b.u32((uint32_t)stts.size());
              for (auto& [cnt, delta] : stts) { b.u32(cnt); b.u32(delta); }
              stts_b.close();
            }
            // ctts only if any nonzero offset in range
            bool any_cts = false;
            for (size_t i = 0; i < m; i++)
              if (d->cts[lo + i] != 0) any_cts = true;
            if (any_cts) {
              BoxW ctts(b, "ctts");
              b.u32(0);
              // run-length encode
              std::vector<std::pair<uint32_t, int32_t>> runs;
              for (size_t i = 0; i < m; i++) {
                int32_t off = d->cts[lo + i];
                if (!runs.empty() && runs.back().second == off)
                  runs.back().first++;
                else
                  runs.push_back({1, off});
              }
              b.u32((uint32_t)runs.size());
              for (auto& [cnt, off] : runs) { b.u32(cnt); b.u32((uint32_t)off); }
              ctts.close();
            }
            if (!d->sync.empty()) {
              std::vector<uint32_t> ss;
              for (uint32_t s : d->sync)
                if (s >= lo + 1 && s <= (uint32_t)(hi + 1)) ss.push_back(s - (uint32_t)lo);
              BoxW stss(b, "stss");
              b.u32(0); b.u32((uint32_t)ss.size());
              for (uint32_t s : ss) b.u32(s);
              stss.close();
            }
            {
              BoxW stsz(b, "stsz");
              b.u32(0); b.u32(0); b.u32((uint32_t)m);
              for (uint32_t sz : sizes) b.u32(sz);
              stsz.close();
            }
            {
              BoxW stsc(b, "stsc");
              b.u32(0); b.u32(1); b.u32(1); b.u32((uint32_t)m); b.u32(1);
              stsc.close();
            }
            {
              BoxW stco(b, "stco");
              b.u32(0); b.u32(1); b.u32(chunk_off);
              stco.close();
            }
            stbl.close();
          }
          minf.close();
        }
        mdia.close();
      }
      trak.close();
    }
    moov.close();
    return b.v;
  };

  static const uint8_t ftyp[] = {0, 0, 0, 28, 'f', 't', 'y', 'p', 'i', 's', 'o',
                                 'm', 0, 0, 2, 0, 'i', 's', 'o', 'm', 'a', 'v',
                                 'c', '1', 'm', 'p', '4', '1'};
  auto moov0 = build_moov(0);
  uint32_t chunk_off = (uint32_t)(sizeof(ftyp) + moov0.size() + 8);
  auto moov = build_moov(chunk_off);
  size_t total = sizeof(ftyp) + moov.size() + 8 + mdat_payload.v.size();
  uint8_t* buf = (uint8_t*)malloc(total);
  if (!buf) return cc::set_error(CC_ERR_NOMEM, "remux alloc");
  size_t pos = 0;
  memcpy(buf, ftyp, sizeof(ftyp)); pos += sizeof(ftyp);
  memcpy(buf + pos, moov.data(), moov.size()); pos += moov.size();
  uint32_t mdat_sz = (uint32_t)(mdat_payload.v.size() + 8);
  uint8_t hdr[8] = {(uint8_t)(mdat_sz >> 24), (uint8_t)(mdat_sz >> 16),
                    (uint8_t)(mdat_sz >> 8), (uint8_t)mdat_sz, 'm', 'd', 'a', 't'};
  memcpy(buf + pos, hdr, 8); pos += 8;
  memcpy(buf + pos, mdat_payload.v.data(), mdat_payload.v.size());
  *out = buf;
  *out_size = total;
  return CC_OK;
}

}  // extern "C"
