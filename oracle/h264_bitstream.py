"""Minimal H.264 bitstream reader — an INDEPENDENT pin for the demuxer.

Oracle/test infrastructure ONLY (tier contract): nothing in the product
path imports this module.

The mp4 demuxer's PTS contract (oracle/mp4_demux.py and csrc/
cc_demux.cpp) is derived from the container's stts/ctts/elst tables.
This module reads the OTHER authority inside the same file — the H.264
elementary stream the encoder wrote — and exposes:

  * SPS geometry (crop-adjusted width/height) and VUI timing (fps),
    which must agree with the container's stsd dims and stts-derived
    rate;
  * per-access-unit picture order counts (POC, ISO/IEC 14496-10
    §8.2.1.1 type-0 derivation), whose sort defines display order —
    which must equal the ctts-derived presentation order.

Scope: enough of the spec for the reference's AVC fixtures and our
synthetic streams — poc_type 0/2, frame_mbs_only (progressive), single
SPS/PPS.  Anything outside scope raises rather than guessing.

References: ISO/IEC 14496-10 (2020) §7.3.2.1 (SPS), §7.3.2.2 (PPS),
§7.3.3 (slice header), §8.2.1 (POC), Annex E (VUI).
"""

from __future__ import annotations

from dataclasses import dataclass


class BitReader:
    """MSB-first bit reader over an RBSP (emulation bytes removed)."""

    def __init__(self, data: bytes) -> None:
        self.data = data
        self.pos = 0  # bit position

    def u(self, n: int) -> int:
        v = 0
        for _ in range(n):
            byte = self.data[self.pos >> 3]
            v = (v << 1) | ((byte >> (7 - (self.pos & 7))) & 1)
            self.pos += 1
        return v

    def ue(self) -> int:
        zeros = 0
        while self.u(1) == 0:
            zeros += 1
            if zeros > 31:
                raise ValueError("bad exp-Golomb code")
        return (1 << zeros) - 1 + (self.u(zeros) if zeros else 0)

    def se(self) -> int:
        k = self.ue()
        return (k + 1) // 2 if k % 2 else -(k // 2)


def unescape(nal: bytes) -> bytes:
    """Remove 00 00 03 emulation-prevention bytes (§7.4.1.1)."""
    out = bytearray()
    zeros = 0
    for b in nal:
        if zeros >= 2 and b == 3:
            zeros = 0
            continue  # drop the emulation byte
        out.append(b)
        zeros = zeros + 1 if b == 0 else 0
    return bytes(out)


def iter_nals(annexb: bytes):
    """Yield (nal_unit_type, payload-with-header) for AnnexB data."""
    i = 0
    n = len(annexb)
    starts = []
    while i + 3 <= n:
        if annexb[i] == 0 and annexb[i + 1] == 0:
            if annexb[i + 2] == 1:
                starts.append(i + 3)
                i += 3
                continue
            if i + 4 <= n and annexb[i + 2] == 0 and annexb[i + 3] == 1:
                starts.append(i + 4)
                i += 4
                continue
        i += 1
    for j, s in enumerate(starts):
        e = (starts[j + 1] - 3) if j + 1 < len(starts) else n
        # trim the next start code's leading zeros
        while e > s and annexb[e - 1] == 0 and j + 1 < len(starts):
            e -= 1
        nal = annexb[s:e]
        if nal:
            yield nal[0] & 0x1F, nal


@dataclass
class Sps:
    profile_idc: int
    log2_max_frame_num: int
    poc_type: int
    log2_max_poc_lsb: int  # poc_type 0 only
    frame_mbs_only: bool
    width: int
    height: int
    fps: float | None  # from VUI timing, None if absent
    separate_colour_plane: bool


def parse_sps(nal: bytes) -> Sps:
    r = BitReader(unescape(nal[1:]))  # skip nal header byte
    profile_idc = r.u(8)
    r.u(8)  # constraint flags + reserved
    r.u(8)  # level_idc
    r.ue()  # seq_parameter_set_id
    chroma_format_idc = 1
    separate_colour = False
    if profile_idc in (100, 110, 122, 244, 44, 83, 86, 118, 128, 138, 139,
                       134, 135):
        chroma_format_idc = r.ue()
        if chroma_format_idc == 3:
            separate_colour = r.u(1) == 1
        r.ue()  # bit_depth_luma_minus8
        r.ue()  # bit_depth_chroma_minus8
        r.u(1)  # qpprime_y_zero_transform_bypass
        if r.u(1):  # seq_scaling_matrix_present
            for i in range(8 if chroma_format_idc != 3 else 12):
                if r.u(1):  # scaling list present
                    size = 16 if i < 6 else 64
                    last, nxt = 8, 8
                    for _ in range(size):
                        if nxt != 0:
                            nxt = (last + r.se() + 256) % 256
                        last = nxt if nxt != 0 else last
    log2_max_frame_num = r.ue() + 4
    poc_type = r.ue()
    log2_max_poc_lsb = 0
    if poc_type == 0:
        log2_max_poc_lsb = r.ue() + 4
    elif poc_type == 1:
        r.u(1)
        r.se()
        r.se()
        for _ in range(r.ue()):
            r.se()
    r.ue()  # max_num_ref_frames
    r.u(1)  # gaps_in_frame_num_value_allowed
    pic_width_in_mbs = r.ue() + 1
    pic_height_in_map_units = r.ue() + 1
    frame_mbs_only = r.u(1) == 1
    if not frame_mbs_only:
        r.u(1)  # mb_adaptive_frame_field
    r.u(1)  # direct_8x8_inference
    crop_l = crop_r = crop_t = crop_b = 0
    if r.u(1):  # frame_cropping
        crop_l, crop_r, crop_t, crop_b = r.ue(), r.ue(), r.ue(), r.ue()
    fps = None
    if r.u(1):  # vui_parameters_present
        if r.u(1):  # aspect_ratio_info
            if r.u(8) == 255:  # Extended_SAR
                r.u(16)
                r.u(16)
        if r.u(1):  # overscan_info
            r.u(1)
        if r.u(1):  # video_signal_type
            r.u(3)
            r.u(1)
            if r.u(1):  # colour_description
                r.u(24)
        if r.u(1):  # chroma_loc_info
            r.ue()
            r.ue()
        if r.u(1):  # timing_info_present
            num_units_in_tick = r.u(32)
            time_scale = r.u(32)
            if num_units_in_tick:
                # field-based clock: one frame = 2 ticks (E-6/E-7)
                fps = time_scale / (2.0 * num_units_in_tick)
    # crop units for 4:2:0 frame-coded: x2 horizontally, x2 vertically
    sub_w = 2 if chroma_format_idc in (1, 2) else 1
    sub_h = 2 if chroma_format_idc == 1 else 1
    mul_h = 1 if frame_mbs_only else 2
    width = pic_width_in_mbs * 16 - sub_w * (crop_l + crop_r)
    height = (2 - frame_mbs_only) * pic_height_in_map_units * 16 - \
        sub_h * mul_h * (crop_t + crop_b)
    return Sps(profile_idc, log2_max_frame_num, poc_type, log2_max_poc_lsb,
               frame_mbs_only, width, height, fps, separate_colour)


@dataclass
class SliceInfo:
    is_idr: bool
    frame_num: int
    poc_lsb: int  # poc_type 0
    nal_ref_idc: int


def parse_first_slice(nal: bytes, sps: Sps) -> SliceInfo:
    nal_ref_idc = (nal[0] >> 5) & 3
    nal_type = nal[0] & 0x1F
    is_idr = nal_type == 5
    r = BitReader(unescape(nal[1:]))
    r.ue()  # first_mb_in_slice
    r.ue()  # slice_type
    r.ue()  # pic_parameter_set_id
    if sps.separate_colour_plane:
        r.u(2)
    frame_num = r.u(sps.log2_max_frame_num)
    if not sps.frame_mbs_only:
        if r.u(1):  # field_pic_flag
            raise ValueError("field-coded slices out of scope")
    if is_idr:
        r.ue()  # idr_pic_id
    poc_lsb = 0
    if sps.poc_type == 0:
        poc_lsb = r.u(sps.log2_max_poc_lsb)
    return SliceInfo(is_idr, frame_num, poc_lsb, nal_ref_idc)


def access_unit_pocs(packets: list[bytes]) -> list[int]:
    """POC per access unit (decode order); see access_unit_pocs_idr."""
    return [p for p, _ in access_unit_pocs_idr(packets)]


def access_unit_pocs_idr(packets: list[bytes]) -> list[tuple[int, bool]]:
    """(POC, is_idr) per access unit (decode order), §8.2.1.1 type-0
    derivation (or frame_num-based order for poc_type 2).

    POC orders pictures only WITHIN one coded video sequence: it resets
    at every IDR, so display-order comparisons must be made per
    IDR-delimited segment (the is_idr flags mark the boundaries).

    `packets` are AnnexB access units in decode order (cc_demux_packet
    output — SPS/PPS prefixed on sync samples).
    """
    sps: Sps | None = None
    pocs: list[int] = []
    prev_msb = 0
    prev_lsb = 0
    for au in packets:
        slice_nal = None
        for t, nal in iter_nals(au):
            if t == 7:
                sps = parse_sps(nal)
            elif t in (1, 5) and slice_nal is None:
                slice_nal = nal
        if slice_nal is None:
            raise ValueError("access unit without a slice NAL")
        if sps is None:
            raise ValueError("slice before SPS")
        si = parse_first_slice(slice_nal, sps)
        if sps.poc_type == 0:
            max_lsb = 1 << sps.log2_max_poc_lsb
            if si.is_idr:
                prev_msb, prev_lsb = 0, 0
            if si.poc_lsb < prev_lsb and (prev_lsb - si.poc_lsb) >= max_lsb // 2:
                msb = prev_msb + max_lsb
            elif si.poc_lsb > prev_lsb and (si.poc_lsb - prev_lsb) > max_lsb // 2:
                msb = prev_msb - max_lsb
            else:
                msb = prev_msb
            poc = msb + si.poc_lsb
            if si.nal_ref_idc:  # only reference pictures update prev*
                prev_msb, prev_lsb = msb, si.poc_lsb
            pocs.append((poc, si.is_idr))
        elif sps.poc_type == 2:
            # display order == decode order; synthesize increasing POC
            pocs.append((2 * len(pocs), si.is_idr))
        else:
            raise ValueError(f"poc_type {sps.poc_type} out of scope")
    return pocs


def sps_of_packets(packets: list[bytes]) -> Sps:
    for au in packets:
        for t, nal in iter_nals(au):
            if t == 7:
                return parse_sps(nal)
    raise ValueError("no SPS found")
