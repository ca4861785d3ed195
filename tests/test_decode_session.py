"""rocDecode session wiring exercised against a mocked librocdecode.so.

The real librocdecode does not exist in this image, so the
COMPLETE session code in csrc/cc_decode.cpp (parser + lazy decoder +
display queue + surface recycling) is driven here by a mock that
implements the public rocDec* API (tools/mock_rocdecode.cpp, built
against the same rocprofiler-sdk headers).  The mock is injected via
LD_LIBRARY_PATH in a subprocess, since libcchot.so caches the dlopen
probe per process.  CPU-only: the wiring passes device pointers through
without dereferencing them.
"""

from __future__ import annotations

import pathlib
import subprocess
import sys

import pytest

ROOT = pathlib.Path(__file__).resolve().parent.parent
MOCK_SRC = ROOT / "tools" / "mock_rocdecode.cpp"


@pytest.fixture(scope="module")
def mock_dir(tmp_path_factory):
    d = tmp_path_factory.mktemp("mock_rocdecode")
    subprocess.run(
        ["g++", "-shared", "-fPIC", "-D__HIP_PLATFORM_AMD__",
         "-I/opt/rocm/include", str(MOCK_SRC), "-o",
         str(d / "librocdecode.so")],
        check=True,
    )
    return d


DRIVER = r"""
import ctypes
import json
import sys

sys.path.insert(0, ".")
from cosmos_curate_amd import hotpath

lib = hotpath.load()
assert lib.cc_rocdecode_available() == 0, lib.cc_last_error().decode()

mock = ctypes.CDLL("librocdecode.so")  # same handle the session dlopened

s = hotpath.DecodeSession(device=0, codec=0)
# 5 access units, pts in our demuxer units
for i in range(5):
    s.submit(b"\x00\x00\x00\x01\x65" + bytes([i]) * 16, pts=1000 * (i + 1))
frames = s.map_frames(cap=8)
out = {
    "n": len(frames),
    "pts": [f.pts for f in frames],
    "wh": [[f.width, f.height] for f in frames],
    "y": [f.y for f in frames],
    "uv": [f.uv for f in frames],
    "pitch": [f.pitch for f in frames],
    "decoded_before_flush": mock.mock_decoded(),
}
# flush (end of stream) then recycle the mapped surfaces
s.submit(None)
out["eos_seen"] = mock.mock_eos_seen()
s.recycle()
out["reused"] = mock.mock_reused()
s.close()
out["destroyed"] = mock.mock_destroyed()

# error path: HEVC session on the same mock, bad codec id rejected
s2 = hotpath.DecodeSession(device=0, codec=1)
s2.close()
try:
    hotpath.DecodeSession(device=0, codec=7)
    out["bad_codec_rejected"] = False
except RuntimeError:
    out["bad_codec_rejected"] = True

print(json.dumps(out))
"""


def test_session_full_lifecycle_with_mock(mock_dir):
    import json
    import os

    env = dict(os.environ, LD_LIBRARY_PATH=str(mock_dir))
    r = subprocess.run([sys.executable, "-c", DRIVER], cwd=ROOT, env=env,
                       capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, r.stdout + r.stderr
    out = json.loads(r.stdout.strip().splitlines()[-1])
    # 5 packets -> 5 decode calls -> 5 display-order frames, in order
    assert out["n"] == 5
    assert out["decoded_before_flush"] == 5
    assert out["pts"] == [1000, 2000, 3000, 4000, 5000]
    # display crop (48x30 of the 48x32 coded surface) is what callers see
    assert out["wh"] == [[48, 30]] * 5
    # distinct surfaces with the mock's pointer scheme + pitch
    assert len(set(out["y"])) == 5
    assert all(uv == y + 0x8000 for y, uv in zip(out["y"], out["uv"]))
    assert out["pitch"] == [64] * 5
    assert out["eos_seen"] == 1
    # recycle marked every mapped surface for reuse
    assert out["reused"] == 5
    # destroy tore down parser + decoder
    assert out["destroyed"] >= 2
    assert out["bad_codec_rejected"]


def test_without_library_fails_loudly():
    """No librocdecode on the default path: CC_ERR_NO_ROCDECODE (-5),
    never a silent fallback (tier contract)."""
    import ctypes
    import os

    if os.environ.get("LD_LIBRARY_PATH", "").find("mock_rocdecode") >= 0:
        pytest.skip("mock on path")
    from cosmos_curate_amd import hotpath

    lib = hotpath.load()
    if lib.cc_rocdecode_available() == 0:
        pytest.skip("real librocdecode present")
    h = ctypes.c_void_p()
    rc = lib.cc_decode_session_create(0, 0, ctypes.byref(h))
    assert rc == -5  # CC_ERR_NO_ROCDECODE
