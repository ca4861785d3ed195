"""From-scratch MP4 (ISO BMFF) demuxer for PTS extraction (CPU oracle).

Restates the semantics of ``get_video_timestamps``
(/root/reference/cosmos_curate/pipelines/video/utils/decoder_utils.py:230-278):
presentation timestamps of every video packet, in seconds (float32), sorted
ascending.  The reference obtains them via PyAV demux (libavformat); here they
come straight from the container's sample tables, which is where libavformat
reads them too:

    stts (decode deltas) -> per-sample DTS
    ctts (composition offsets) -> PTS = DTS + offset
    elst (edit list) -> presentation shift: libavformat's mov demuxer
        subtracts the first edit's media_time so presentation starts at the
        edit point (x264-style "priming" delay ends up at pts 0).
    mdhd timescale -> seconds

Also exposes sample byte ranges (stsz/stco/stsc) so parity tests can check
the C++ host demuxer's packet slicing, and avcC parsing for AnnexB conversion.

Pure stdlib + numpy; no third-party media code.
"""

from __future__ import annotations

import io
import struct
from dataclasses import dataclass, field

import numpy as np
import numpy.typing as npt


@dataclass
class Mp4Track:
    """One video track's sample tables."""

    timescale: int = 0
    codec: str = ""
    width: int = 0
    height: int = 0
    # per-sample
    sizes: list[int] = field(default_factory=list)
    offsets: list[int] = field(default_factory=list)
    dts: list[int] = field(default_factory=list)
    cts_offset: list[int] = field(default_factory=list)
    sync_samples: list[int] = field(default_factory=list)  # 1-based, empty = all sync
    elst_media_time: int = 0  # first non-empty edit's media_time
    avcc: bytes = b""  # raw AvcDecoderConfigurationRecord (h264)
    hvcc: bytes = b""  # raw HEVCDecoderConfigurationRecord (hevc)

    @property
    def pts(self) -> list[int]:
        return [d + c for d, c in zip(self.dts, self.cts_offset)]

    def pts_seconds_sorted(self) -> npt.NDArray[np.float32]:
        """The get_video_timestamps contract: sorted f32 seconds.

        Matches decoder_utils.py:275-278: each packet's pts * time_base as
        float32, then np.sort.
        """
        shift = self.elst_media_time
        ts = np.array(
            [np.float32(float(p - shift) / float(self.timescale)) for p in self.pts],
            dtype=np.float32,
        )
        return np.sort(ts)


def _iter_boxes(buf: bytes, start: int, end: int):
    pos = start
    while pos + 8 <= end:
        size, typ = struct.unpack_from(">I4s", buf, pos)
        hdr = 8
        if size == 1:
            size = struct.unpack_from(">Q", buf, pos + 8)[0]
            hdr = 16
        elif size == 0:
            size = end - pos
        if size < hdr or pos + size > end:
            break
        yield typ.decode("latin1"), pos + hdr, pos + size
        pos += size


def parse_mp4(data: bytes) -> list[Mp4Track]:
    """Parse an MP4/MOV byte stream into per-video-track sample tables."""
    tracks: list[Mp4Track] = []
    for typ, b, e in _iter_boxes(data, 0, len(data)):
        if typ == "moov":
            for t2, b2, e2 in _iter_boxes(data, b, e):
                if t2 == "trak":
                    trk = _parse_trak(data, b2, e2)
                    if trk is not None:
                        tracks.append(trk)
    return tracks


def _parse_trak(buf: bytes, start: int, end: int) -> Mp4Track | None:
    trk = Mp4Track()
    is_video = False
    for typ, b, e in _iter_boxes(buf, start, end):
        if typ == "edts":
            for t2, b2, e2 in _iter_boxes(buf, b, e):
                if t2 == "elst":
                    trk.elst_media_time = _parse_elst(buf, b2)
        elif typ == "mdia":
            for t2, b2, e2 in _iter_boxes(buf, b, e):
                if t2 == "mdhd":
                    ver = buf[b2]
                    trk.timescale = struct.unpack_from(
                        ">I", buf, b2 + (20 if ver == 1 else 12)
                    )[0]
                elif t2 == "hdlr":
                    if buf[b2 + 8 : b2 + 12] == b"vide":
                        is_video = True
                elif t2 == "minf":
                    for t3, b3, e3 in _iter_boxes(buf, b2, e2):
                        if t3 == "stbl":
                            _parse_stbl(buf, b3, e3, trk)
    return trk if is_video and trk.timescale > 0 else None


def _parse_elst(buf: bytes, b: int) -> int:
    ver = buf[b]
    n = struct.unpack_from(">I", buf, b + 4)[0]
    pos = b + 8
    for _ in range(n):
        if ver == 1:
            dur, media_time = struct.unpack_from(">Qq", buf, pos)
            pos += 20
        else:
            dur, media_time = struct.unpack_from(">Ii", buf, pos)
            pos += 12
        if media_time != -1:  # first non-empty edit
            return int(media_time)
    return 0


def _parse_stbl(buf: bytes, start: int, end: int, trk: Mp4Track) -> None:
    stsc_entries: list[tuple[int, int]] = []  # (first_chunk, samples_per_chunk)
    chunk_offsets: list[int] = []
    for typ, b, e in _iter_boxes(buf, start, end):
        if typ == "stsd":
            n = struct.unpack_from(">I", buf, b + 4)[0]
            pos = b + 8
            for _ in range(n):
                esize, efmt = struct.unpack_from(">I4s", buf, pos)
                trk.codec = efmt.decode("latin1")
                if trk.codec in ("avc1", "hvc1", "hev1", "avc3"):
                    trk.width, trk.height = struct.unpack_from(">HH", buf, pos + 32)
                    for t2, b2, e2 in _iter_boxes(buf, pos + 86, pos + esize):
                        if t2 == "avcC":
                            trk.avcc = bytes(buf[b2:e2])
                        elif t2 == "hvcC":
                            trk.hvcc = bytes(buf[b2:e2])
                pos += esize
        elif typ == "stts":
            n = struct.unpack_from(">I", buf, b + 4)[0]
            t = 0
            pos = b + 8
            for _ in range(n):
                cnt, delta = struct.unpack_from(">II", buf, pos)
                pos += 8
                for _ in range(cnt):
                    trk.dts.append(t)
                    t += delta
        elif typ == "ctts":
            n = struct.unpack_from(">I", buf, b + 4)[0]
            pos = b + 8
            for _ in range(n):
                cnt, off = struct.unpack_from(">Ii", buf, pos)  # signed ok for v1
                pos += 8
                trk.cts_offset.extend([off] * cnt)
        elif typ == "stss":
            n = struct.unpack_from(">I", buf, b + 4)[0]
            trk.sync_samples = list(
                struct.unpack_from(f">{n}I", buf, b + 8)
            )
        elif typ == "stsz":
            sample_size, n = struct.unpack_from(">II", buf, b + 4)
            if sample_size:
                trk.sizes = [sample_size] * n
            else:
                trk.sizes = list(struct.unpack_from(f">{n}I", buf, b + 12))
        elif typ == "stsc":
            n = struct.unpack_from(">I", buf, b + 4)[0]
            pos = b + 8
            for _ in range(n):
                first, spc, _desc = struct.unpack_from(">III", buf, pos)
                pos += 12
                stsc_entries.append((first, spc))
        elif typ in ("stco", "co64"):
            n = struct.unpack_from(">I", buf, b + 4)[0]
            fmt = ">Q" if typ == "co64" else ">I"
            sz = 8 if typ == "co64" else 4
            chunk_offsets = [
                struct.unpack_from(fmt, buf, b + 8 + i * sz)[0] for i in range(n)
            ]

    if not trk.cts_offset:
        trk.cts_offset = [0] * len(trk.dts)
    # expand chunk map -> per-sample file offsets
    if chunk_offsets and stsc_entries and trk.sizes:
        per_chunk: list[int] = []
        for i, (first, spc) in enumerate(stsc_entries):
            last = (
                stsc_entries[i + 1][0] - 1
                if i + 1 < len(stsc_entries)
                else len(chunk_offsets)
            )
            per_chunk.extend([spc] * (last - first + 1))
        si = 0
        for ci, coff in enumerate(chunk_offsets):
            pos = coff
            for _ in range(per_chunk[ci] if ci < len(per_chunk) else 0):
                if si >= len(trk.sizes):
                    break
                trk.offsets.append(pos)
                pos += trk.sizes[si]
                si += 1


def get_video_timestamps(data: bytes) -> npt.NDArray[np.float32]:
    """Oracle counterpart of decoder_utils.get_video_timestamps (:230-278)."""
    tracks = parse_mp4(data)
    if not tracks:
        msg = "no video track found"
        raise ValueError(msg)
    return tracks[0].pts_seconds_sorted()


def annexb_packets(data: bytes, track: Mp4Track) -> list[bytes]:
    """Per-sample AnnexB bitstream (length-prefixed NALs -> start codes).

    Mirrors what PyNvDemuxer hands to NVDEC (nvcodec_utils.py:224); used as
    the golden reference for the C++ host demuxer's packet output.
    """
    if track.avcc:
        nal_len = (track.avcc[4] & 0x03) + 1
        # parameter sets from avcC
        prefix = b""
        pos = 5
        num_sps = track.avcc[pos] & 0x1F
        pos += 1
        for _ in range(num_sps):
            ln = struct.unpack_from(">H", track.avcc, pos)[0]
            prefix += b"\x00\x00\x00\x01" + track.avcc[pos + 2 : pos + 2 + ln]
            pos += 2 + ln
        num_pps = track.avcc[pos]
        pos += 1
        for _ in range(num_pps):
            ln = struct.unpack_from(">H", track.avcc, pos)[0]
            prefix += b"\x00\x00\x00\x01" + track.avcc[pos + 2 : pos + 2 + ln]
            pos += 2 + ln
    elif track.hvcc:
        # HEVCDecoderConfigurationRecord (ISO 14496-15 §8.3.3.1):
        # 22-byte header, numOfArrays, arrays of (type, count, nalus)
        h = track.hvcc
        nal_len = (h[21] & 0x03) + 1
        prefix = b""
        pos = 23
        for _ in range(h[22]):
            cnt = struct.unpack_from(">H", h, pos + 1)[0]
            pos += 3
            for _ in range(cnt):
                ln = struct.unpack_from(">H", h, pos)[0]
                prefix += b"\x00\x00\x00\x01" + h[pos + 2 : pos + 2 + ln]
                pos += 2 + ln
    else:
        msg = "avcC/hvcC missing"
        raise ValueError(msg)

    out: list[bytes] = []
    sync = set(track.sync_samples) if track.sync_samples else None
    for i, (off, size) in enumerate(zip(track.offsets, track.sizes)):
        sample = data[off : off + size]
        b = io.BytesIO()
        if (sync is None or (i + 1) in sync) and prefix:
            b.write(prefix)
        p = 0
        while p + nal_len <= len(sample):
            ln = int.from_bytes(sample[p : p + nal_len], "big")
            p += nal_len
            b.write(b"\x00\x00\x00\x01")
            b.write(sample[p : p + ln])
            p += ln
        out.append(b.getvalue())
    return out
