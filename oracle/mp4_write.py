"""Minimal MP4 writer for demux-test fixtures (CPU oracle infrastructure).

Produces tiny ISO-BMFF files with controlled stts/ctts/elst/stss tables and
arbitrary (fake) sample payloads.  Demux never decodes, so parser tests need
valid *tables*, not valid H.264 -- this lets us commit a few-KB fixture that
exercises B-frame reordering and edit-list shifts without shipping real
media.  Not part of the product path.
"""

from __future__ import annotations

import struct


def _box(typ: bytes, payload: bytes) -> bytes:
    return struct.pack(">I4s", 8 + len(payload), typ) + payload


def _full(typ: bytes, ver: int, flags: int, payload: bytes) -> bytes:
    return _box(typ, struct.pack(">B3s", ver, flags.to_bytes(3, "big")) + payload)


def write_mp4(
    sample_sizes: list[int],
    stts: list[tuple[int, int]],
    ctts: list[tuple[int, int]] | None,
    timescale: int,
    *,
    elst_media_time: int | None = None,
    sync_samples: list[int] | None = None,
    width: int = 64,
    height: int = 64,
    payload_byte: int = 0xAB,
    codec: str = "h264",
    samples_per_chunk: int | None = None,
    use_co64: bool = False,
) -> bytes:
    """Build an MP4 with one avc1 (or hvc1) video track and fake samples."""
    n = len(sample_sizes)
    mdat_payload = bytes([payload_byte]) * sum(sample_sizes)
    # layout: ftyp, moov, mdat.  Compute moov size by building it with a
    # placeholder chunk offset first, then patching (single chunk).
    ftyp = _box(b"ftyp", b"isom" + struct.pack(">I", 512) + b"isomiso2avc1mp41")

    fake_avcc = bytes(
        [1, 0x42, 0xC0, 0x1E, 0xFF, 0xE1, 0, 4, 0x67, 0x42, 0xC0, 0x1E, 1, 0, 2, 0x68, 0xCE]
    )
    # HEVCDecoderConfigurationRecord: 22-byte header (byte 21 low bits =
    # lengthSizeMinusOne=3 -> 4-byte NAL lengths), numOfArrays=3, then one
    # fake VPS(32)/SPS(33)/PPS(34) NAL each (ISO 14496-15 §8.3.3.1)
    fake_hvcc = (
        bytes([1]) + bytes(20) + bytes([0x03, 3])
        + bytes([0xA0, 0, 1, 0, 3, 0x40, 1, 2])       # VPS array, 1 nal len 3
        + bytes([0xA1, 0, 1, 0, 4, 0x42, 1, 2, 3])    # SPS array, 1 nal len 4
        + bytes([0xA2, 0, 1, 0, 2, 0x44, 1])          # PPS array, 1 nal len 2
    )
    entry_type, cfg_box = (
        (b"hvc1", _box(b"hvcC", fake_hvcc)) if codec == "hevc"
        else (b"avc1", _box(b"avcC", fake_avcc))
    )
    avc1 = _box(
        entry_type,
        b"\x00" * 6
        + struct.pack(">H", 1)
        + b"\x00" * 16
        + struct.pack(">HH", width, height)
        + struct.pack(">II", 0x00480000, 0x00480000)
        + b"\x00" * 4
        + struct.pack(">H", 1)
        + b"\x00" * 32
        + struct.pack(">H", 0x18)
        + struct.pack(">h", -1)
        + cfg_box,
    )
    stsd = _full(b"stsd", 0, 0, struct.pack(">I", 1) + avc1)
    stts_b = _full(
        b"stts", 0, 0,
        struct.pack(">I", len(stts)) + b"".join(struct.pack(">II", c, d) for c, d in stts),
    )
    ctts_b = b""
    if ctts is not None:
        ctts_b = _full(
            b"ctts", 0, 0,
            struct.pack(">I", len(ctts)) + b"".join(struct.pack(">Ii", c, o) for c, o in ctts),
        )
    stss_b = b""
    if sync_samples is not None:
        stss_b = _full(
            b"stss", 0, 0,
            struct.pack(">I", len(sync_samples))
            + b"".join(struct.pack(">I", s) for s in sync_samples),
        )
    stsz = _full(
        b"stsz", 0, 0,
        struct.pack(">II", 0, n) + b"".join(struct.pack(">I", s) for s in sample_sizes),
    )
    spc = samples_per_chunk if samples_per_chunk else n
    n_chunks = (n + spc - 1) // spc
    stsc = _full(b"stsc", 0, 0, struct.pack(">I", 1) + struct.pack(">III", 1, spc, 1))
    co_type = b"co64" if use_co64 else b"stco"
    co_fmt = ">Q" if use_co64 else ">I"
    stco_placeholder = _full(
        co_type, 0, 0,
        struct.pack(">I", n_chunks) + struct.pack(co_fmt, 0) * n_chunks)

    def build_moov(chunk_off: int) -> bytes:
        # chunk c starts at the sum of sample sizes before it (contiguous mdat)
        offs, acc = [], chunk_off
        for c in range(n_chunks):
            offs.append(acc)
            acc += sum(sample_sizes[c * spc:(c + 1) * spc])
        stco = _full(co_type, 0, 0,
                     struct.pack(">I", n_chunks)
                     + b"".join(struct.pack(co_fmt, o) for o in offs))
        stbl = _box(b"stbl", stsd + stts_b + ctts_b + stss_b + stsz + stsc + stco)
        total_dur = sum(c * d for c, d in stts)
        mdhd = _full(
            b"mdhd", 0, 0, struct.pack(">IIIIHH", 0, 0, timescale, total_dur, 0x55C4, 0)
        )
        hdlr = _full(b"hdlr", 0, 0, struct.pack(">I", 0) + b"vide" + b"\x00" * 12 + b"v\x00")
        vmhd = _full(b"vmhd", 0, 1, b"\x00" * 8)
        dref = _full(b"dref", 0, 0, struct.pack(">I", 1) + _full(b"url ", 0, 1, b""))
        dinf = _box(b"dinf", dref)
        minf = _box(b"minf", vmhd + dinf + stbl)
        mdia = _box(b"mdia", mdhd + hdlr + minf)
        tkhd = _full(
            b"tkhd", 0, 7,
            struct.pack(">IIII", 0, 0, 1, 0)
            + struct.pack(">I", total_dur)
            + b"\x00" * 16
            + struct.pack(">9i", 0x10000, 0, 0, 0, 0x10000, 0, 0, 0, 0x40000000)
            + struct.pack(">II", width << 16, height << 16),
        )
        edts = b""
        if elst_media_time is not None:
            elst = _full(
                b"elst", 0, 0,
                struct.pack(">I", 1) + struct.pack(">Ii", total_dur, elst_media_time)
                + struct.pack(">HH", 1, 0),
            )
            edts = _box(b"edts", elst)
        trak = _box(b"trak", tkhd + edts + mdia)
        mvhd = _full(
            b"mvhd", 0, 0,
            struct.pack(">IIII", 0, 0, timescale, total_dur)
            + struct.pack(">IHHII", 0x10000, 0x100, 0, 0, 0)
            + struct.pack(">9i", 0x10000, 0, 0, 0, 0x10000, 0, 0, 0, 0x40000000)
            + b"\x00" * 24
            + struct.pack(">I", 2),
        )
        return _box(b"moov", mvhd + trak)

    moov_size = len(build_moov(0))
    chunk_off = len(ftyp) + moov_size + 8  # mdat payload starts after its header
    moov = build_moov(chunk_off)
    assert len(moov) == moov_size
    mdat = _box(b"mdat", mdat_payload)
    return ftyp + moov + mdat
