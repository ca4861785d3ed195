"""Raw-NV12 clip payload — the pluggable codec backend's null codec.

The hot path downstream of hardware decode consumes NV12 surfaces in HBM
(SURVEY.md §8 row a5).  This ROCm image ships no librocdecode (see
csrc/cc_decode.cpp), so kernel-path tests and the bench feed the pipeline
clips whose ``encoded_data`` is a raw NV12 payload instead of H.264
(SURVEY.md §7 hard part (d): "design the ABI so the codec is a pluggable
backend (rocDecode | null-raw) and carry raw-frame fixtures").

Payload layout (little-endian):
    magic  b"CCNV12RAW\\0"  (10 bytes)
    u16 version=1
    u32 n_frames, height, width, fps_num, fps_den
    then per frame: Y plane (h*w bytes) + interleaved UV plane (h/2*w bytes)
"""

from __future__ import annotations

import struct

import numpy as np
import numpy.typing as npt

MAGIC = b"CCNV12RAW\x00"
_HDR = struct.Struct("<10sHIIIII")
HEADER_SIZE = _HDR.size


def pack_header(n: int, h: int, w: int, fps_num: int, fps_den: int = 1) -> bytes:
    """The 32-byte payload header alone (for zero-copy body slicing)."""
    return _HDR.pack(MAGIC, 1, n, h, w, fps_num, fps_den)


def encode_raw_nv12(
    y: npt.NDArray[np.uint8], uv: npt.NDArray[np.uint8], fps_num: int, fps_den: int = 1
) -> bytes:
    """Pack (N,H,W) Y + (N,H/2,W) interleaved-UV planes into a payload."""
    n, h, w = y.shape
    assert uv.shape == (n, h // 2, w), (y.shape, uv.shape)
    hdr = _HDR.pack(MAGIC, 1, n, h, w, fps_num, fps_den)
    frames = np.concatenate(
        [np.concatenate([y[i].reshape(-1), uv[i].reshape(-1)]) for i in range(n)]
    )
    return hdr + frames.tobytes()


def is_raw_nv12(data: bytes | npt.NDArray[np.uint8]) -> bool:
    b = bytes(data[: len(MAGIC)])
    return b == MAGIC


def parse_header(data: bytes) -> tuple[int, int, int, float]:
    """Returns (n_frames, height, width, fps)."""
    magic, ver, n, h, w, num, den = _HDR.unpack_from(bytes(data[: _HDR.size]))
    assert magic == MAGIC and ver == 1
    return n, h, w, num / den


def frame_planes(
    data: bytes, indices: npt.NDArray[np.int32]
) -> tuple[npt.NDArray[np.uint8], npt.NDArray[np.uint8]]:
    """Host gather of selected frames' (Y, UV) planes from the payload."""
    n, h, w, _ = parse_header(data)
    fsz = h * w + (h // 2) * w
    body = np.frombuffer(data, dtype=np.uint8, offset=_HDR.size)
    ys = np.empty((len(indices), h, w), dtype=np.uint8)
    uvs = np.empty((len(indices), h // 2, w), dtype=np.uint8)
    for j, i in enumerate(indices.tolist()):
        base = i * fsz
        ys[j] = body[base : base + h * w].reshape(h, w)
        uvs[j] = body[base + h * w : base + fsz].reshape(h // 2, w)
    return ys, uvs


def timestamps(data: bytes) -> npt.NDArray[np.float32]:
    """Synthesized PTS grid i/fps, float32 (the null codec's demux)."""
    n, _, _, fps = parse_header(data)
    return (np.arange(n, dtype=np.float32) / np.float32(fps)).astype(np.float32)


def make_synthetic_clip(
    n_frames: int, height: int, width: int, fps: int, seed: int
) -> bytes:
    """Seeded moving-gradient + noise NV12 clip (BASELINE.md corpus recipe)."""
    rng = np.random.default_rng(seed)
    yy, xx = np.mgrid[0:height, 0:width]
    t = np.arange(n_frames)[:, None, None]
    y = ((xx[None] * 255 // max(width, 1) + yy[None] // 2 + t * 7) % 256).astype(np.uint8)
    y = np.clip(
        y.astype(np.int16) + rng.integers(-12, 13, size=y.shape, dtype=np.int16), 0, 255
    ).astype(np.uint8)
    uv = rng.integers(96, 160, size=(n_frames, height // 2, width), dtype=np.uint8).astype(
        np.uint8
    )
    return encode_raw_nv12(y, uv, fps)
