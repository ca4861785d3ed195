"""ctypes binding to libcchot.so — the product side of the C ABI.

Declarations mirror include/cc_hotpath.h one-to-one.  Host entry points
(demux) work anywhere; device entry points require a GPU and FAIL LOUDLY
when the extension or the device is missing — there is no CPU fallback in
the product path (tier contract; see also csrc/cc_decode.cpp for the
rocDecode seam).
"""

from __future__ import annotations

import ctypes
import pathlib

import numpy as np
import numpy.typing as npt

_LIB_PATH = pathlib.Path(__file__).resolve().parent / "lib" / "libcchot.so"
_lib: ctypes.CDLL | None = None


class HotpathUnavailableError(RuntimeError):
    """libcchot.so missing or unusable — the product GPU path must not run."""


def _declare(lib: ctypes.CDLL) -> None:
    c = ctypes
    lib.cc_last_error.restype = c.c_char_p
    lib.cc_hip_available.restype = c.c_int
    lib.cc_rocdecode_available.restype = c.c_int

    lib.cc_demux_open.argtypes = [c.c_char_p, c.c_size_t, c.POINTER(c.c_void_p)]
    lib.cc_demux_probe.argtypes = [c.c_void_p, c.c_void_p]
    lib.cc_demux_timestamps.argtypes = [
        c.c_void_p, c.c_void_p, c.c_size_t, c.POINTER(c.c_size_t)]
    lib.cc_demux_packet.argtypes = [
        c.c_void_p, c.c_size_t, c.POINTER(c.c_void_p), c.POINTER(c.c_size_t),
        c.POINTER(c.c_int64), c.POINTER(c.c_int32)]
    lib.cc_demux_close.argtypes = [c.c_void_p]
    lib.cc_demux_remux_clip.argtypes = [
        c.c_void_p, c.c_double, c.c_double, c.POINTER(c.c_void_p),
        c.POINTER(c.c_size_t)]
    lib.cc_buffer_free.argtypes = [c.c_void_p]

    lib.cc_decode_session_create.argtypes = [
        c.c_int, c.c_int32, c.POINTER(c.c_void_p)]
    lib.cc_decode_submit.argtypes = [
        c.c_void_p, c.c_char_p, c.c_size_t, c.c_int64]
    lib.cc_decode_map_frames.argtypes = [
        c.c_void_p, c.c_void_p, c.c_size_t, c.POINTER(c.c_size_t)]
    lib.cc_decode_recycle.argtypes = [c.c_void_p]
    lib.cc_decode_destroy.argtypes = [c.c_void_p]

    lib.cc_malloc.argtypes = [c.POINTER(c.c_void_p), c.c_size_t]
    lib.cc_free.argtypes = [c.c_void_p]
    lib.cc_memcpy_h2d.argtypes = [c.c_void_p, c.c_void_p, c.c_size_t, c.c_uint64]
    lib.cc_memcpy_d2h.argtypes = [c.c_void_p, c.c_void_p, c.c_size_t, c.c_uint64]
    lib.cc_stream_sync.argtypes = [c.c_uint64]

    lib.cc_nv12_to_rgb.argtypes = [
        c.c_void_p, c.c_void_p, c.c_int, c.c_int, c.c_int, c.c_size_t,
        c.c_void_p, c.c_uint64]
    lib.cc_nv12_to_rgb_resize.argtypes = [
        c.c_void_p, c.c_void_p, c.c_int, c.c_int, c.c_int, c.c_size_t,
        c.c_void_p, c.c_int, c.c_int, c.c_uint64]
    lib.cc_resize_bilinear_u8.argtypes = [
        c.c_void_p, c.c_int, c.c_int, c.c_int, c.c_void_p, c.c_int, c.c_int,
        c.c_uint64]
    lib.cc_resize_bicubic_u8.argtypes = lib.cc_resize_bilinear_u8.argtypes
    lib.cc_clip_preprocess.argtypes = [
        c.c_void_p, c.c_int, c.c_int, c.c_int, c.POINTER(c.c_float),
        c.POINTER(c.c_float), c.c_void_p, c.c_int, c.c_uint64]
    lib.cc_clip_preprocess_patches.argtypes = [
        c.c_void_p, c.c_int, c.c_int, c.c_int, c.c_int, c.c_int,
        c.POINTER(c.c_float), c.POINTER(c.c_float), c.c_void_p, c.c_uint64]
    lib.cc_gather_frames_u8.argtypes = [
        c.c_void_p, c.c_int, c.c_size_t, c.c_void_p, c.c_void_p, c.c_int,
        c.c_int, c.c_void_p, c.c_uint64]
    lib.cc_gemm_bf16.argtypes = [
        c.c_void_p, c.c_void_p, c.c_void_p, c.c_int64, c.c_int64, c.c_int64,
        c.c_void_p, c.c_int, c.c_uint64]
    lib.cc_gemm_bf16_ex.argtypes = [
        c.c_void_p, c.c_void_p, c.c_void_p, c.c_int64, c.c_int64, c.c_int64,
        c.c_void_p, c.c_int, c.c_int, c.c_void_p, c.c_uint64]

    lib.cc_attn_small.argtypes = [
        c.c_void_p, c.c_void_p, c.c_int64, c.c_int, c.c_int, c.c_int,
        c.c_float, c.c_uint64]
    lib.cc_attn_mid.argtypes = lib.cc_attn_small.argtypes
    lib.cc_attn_flash.argtypes = lib.cc_attn_small.argtypes
    lib.cc_layernorm_bf16.argtypes = [
        c.c_void_p, c.c_void_p, c.c_void_p, c.c_void_p, c.c_int64, c.c_int64,
        c.c_float, c.c_uint64]
    lib.cc_embed_assemble_ln.argtypes = [
        c.c_void_p, c.c_void_p, c.c_void_p, c.c_void_p, c.c_void_p,
        c.c_void_p, c.c_int64, c.c_int64, c.c_int64, c.c_float, c.c_uint64]
    lib.cc_pairwise_max_earlier.argtypes = [
        c.c_void_p, c.c_int64, c.c_int64, c.c_void_p, c.c_void_p, c.c_uint64]

    lib.cc_timing_enable.argtypes = [c.c_int]
    lib.cc_timing_report.argtypes = [
        c.c_char_p, c.POINTER(c.c_double), c.POINTER(c.c_int64)]


def load() -> ctypes.CDLL:
    """dlopen libcchot.so (idempotent)."""
    global _lib
    if _lib is None:
        if not _LIB_PATH.exists():
            raise HotpathUnavailableError(
                f"{_LIB_PATH} not built — run python -m cosmos_curate_amd.build"
            )
        _lib = ctypes.CDLL(str(_LIB_PATH))
        _declare(_lib)
    return _lib


def available() -> bool:
    try:
        load()
    except HotpathUnavailableError:
        return False
    return True


def require_gpu() -> ctypes.CDLL:
    """The product GPU path gate: extension present AND a HIP device visible."""
    lib = load()
    rc = lib.cc_hip_available()
    if rc != 0:
        raise HotpathUnavailableError(
            f"HIP device unavailable: {lib.cc_last_error().decode()}"
        )
    return lib


def check(rc: int) -> None:
    if rc != 0:
        lib = load()
        raise RuntimeError(f"cc error {rc}: {lib.cc_last_error().decode()}")


class VideoInfo(ctypes.Structure):
    _fields_ = [
        ("width", ctypes.c_uint32),
        ("height", ctypes.c_uint32),
        ("timescale", ctypes.c_uint32),
        ("num_samples", ctypes.c_uint32),
        ("num_sync_samples", ctypes.c_uint32),
        ("codec", ctypes.c_int32),
        ("duration_s", ctypes.c_double),
        ("avg_fps", ctypes.c_double),
    ]


class Nv12Frame(ctypes.Structure):
    """cc_nv12_frame: a mapped VCN NV12 surface (device pointers)."""

    _fields_ = [
        ("y", ctypes.c_void_p),
        ("uv", ctypes.c_void_p),
        ("pitch", ctypes.c_size_t),
        ("pts", ctypes.c_int64),
        ("width", ctypes.c_uint32),
        ("height", ctypes.c_uint32),
    ]


class DecodeSession:
    """RAII wrapper over cc_decode_* (the VCN/rocDecode seam,
    nvcodec_utils.py:199-313 NvVideoDecoder counterpart).

    Raises on construction with CC_ERR_NO_ROCDECODE when librocdecode is
    absent — loud, no CPU fallback.  Mapped surfaces stay valid until
    ``recycle()``; consume (launch + synchronize) first.
    """

    def __init__(self, device: int = 0, codec: int = 0) -> None:
        lib = load()
        h = ctypes.c_void_p()
        check(lib.cc_decode_session_create(device, codec, ctypes.byref(h)))
        self._lib = lib
        self._h: ctypes.c_void_p | None = h

    def submit(self, packet: bytes | None, pts: int = 0) -> None:
        """One AnnexB access unit; None flushes (end of stream)."""
        if packet is None:
            check(self._lib.cc_decode_submit(self._h, None, 0, 0))
        else:
            check(self._lib.cc_decode_submit(self._h, packet, len(packet), pts))

    def map_frames(self, cap: int = 32) -> list[Nv12Frame]:
        arr = (Nv12Frame * cap)()
        n = ctypes.c_size_t()
        check(self._lib.cc_decode_map_frames(
            self._h, ctypes.byref(arr), cap, ctypes.byref(n)))
        return [arr[i] for i in range(n.value)]

    def recycle(self) -> None:
        check(self._lib.cc_decode_recycle(self._h))

    def close(self) -> None:
        if self._h is not None:
            self._lib.cc_decode_destroy(self._h)
            self._h = None

    def __del__(self) -> None:  # pragma: no cover - GC timing
        try:
            self.close()
        except Exception:
            pass


class Demuxer:
    """RAII wrapper over cc_demux_* (host-only, no GPU needed)."""

    def __init__(self, data: bytes) -> None:
        lib = load()
        h = ctypes.c_void_p()
        check(lib.cc_demux_open(data, len(data), ctypes.byref(h)))
        self._lib = lib
        self._h: ctypes.c_void_p | None = h

    def probe(self) -> VideoInfo:
        info = VideoInfo()
        check(self._lib.cc_demux_probe(self._h, ctypes.byref(info)))
        return info

    def timestamps(self) -> npt.NDArray[np.float32]:
        n = ctypes.c_size_t()
        check(self._lib.cc_demux_timestamps(self._h, None, 0, ctypes.byref(n)))
        out = np.empty(n.value, dtype=np.float32)
        check(
            self._lib.cc_demux_timestamps(
                self._h, out.ctypes.data_as(ctypes.c_void_p), n.value, ctypes.byref(n)
            )
        )
        return out

    def packet(self, index: int) -> tuple[bytes, int, bool]:
        p = ctypes.c_void_p()
        sz = ctypes.c_size_t()
        pts = ctypes.c_int64()
        kf = ctypes.c_int32()
        check(
            self._lib.cc_demux_packet(
                self._h, index, ctypes.byref(p), ctypes.byref(sz),
                ctypes.byref(pts), ctypes.byref(kf),
            )
        )
        return ctypes.string_at(p, sz.value), pts.value, bool(kf.value)

    def remux_clip(self, start_s: float, end_s: float) -> bytes:
        """Sample-exact stream-copy of the span into a standalone MP4."""
        buf = ctypes.c_void_p()
        sz = ctypes.c_size_t()
        check(
            self._lib.cc_demux_remux_clip(
                self._h, start_s, end_s, ctypes.byref(buf), ctypes.byref(sz)
            )
        )
        try:
            return ctypes.string_at(buf, sz.value)
        finally:
            self._lib.cc_buffer_free(buf)

    def close(self) -> None:
        if self._h is not None:
            self._lib.cc_demux_close(self._h)
            self._h = None

    def __enter__(self) -> "Demuxer":
        return self

    def __exit__(self, *exc) -> None:
        self.close()

    def __del__(self) -> None:
        try:
            self.close()
        except Exception:
            pass


def timing_enable(on: bool = True) -> None:
    lib = load()
    check(lib.cc_timing_enable(1 if on else 0))
    if on:
        check(lib.cc_timing_reset())


def timing_report(kernel: str) -> tuple[float, int]:
    """(total device ms, launch count) accumulated for `kernel`."""
    lib = load()
    ms = ctypes.c_double()
    cnt = ctypes.c_int64()
    check(lib.cc_timing_report(kernel.encode(), ctypes.byref(ms), ctypes.byref(cnt)))
    return ms.value, cnt.value
