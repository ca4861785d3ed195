"""Generate TransNetV2 golden vectors from the REFERENCE implementation.

Runs only in the dev container (needs /root/reference).  Exec-loads the
reference's pure-torch `_TransNetV2` (models/transnetv2.py:39) with its two
repo-internal imports stubbed (ModelInterface / model_utils — neither
touches the arithmetic), loads the product's deterministic stand-in
weights into it, and records the forward output on a seeded input:

    tests/golden/transnetv2_golden.npz: input (1,100,27,48,3) u8,
                                        output (100,) f32

The product reimplementation (cosmos_curate_amd/models/transnetv2.py) must
reproduce this bit-close (<=1e-5) — pinning the rebuild against the
reference's own network arithmetic.

    python -m oracle.gen_transnet_golden
"""

from __future__ import annotations

import importlib.util
import pathlib
import sys
import types

import numpy as np
import torch

GOLDEN = pathlib.Path(__file__).resolve().parent.parent / "tests" / "golden"
REF = "/root/reference/cosmos_curate/models/transnetv2.py"


def load_reference_transnet():
    mi_mod = types.ModuleType("cosmos_curate.core.interfaces.model_interface")

    class ModelInterface:  # stub: reference class only subclasses it
        pass

    mi_mod.ModelInterface = ModelInterface
    mu_mod = types.ModuleType("cosmos_curate.core.utils.model.model_utils")
    mu_mod.get_local_dir_for_weights_name = lambda n: pathlib.Path("/nonexistent")
    for name in [
        "cosmos_curate", "cosmos_curate.core", "cosmos_curate.core.interfaces",
        "cosmos_curate.core.utils", "cosmos_curate.core.utils.model",
    ]:
        sys.modules.setdefault(name, types.ModuleType(name))
    sys.modules["cosmos_curate.core.interfaces.model_interface"] = mi_mod
    sys.modules["cosmos_curate.core.utils.model.model_utils"] = mu_mod
    spec = importlib.util.spec_from_file_location("ref_transnetv2", REF)
    m = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(m)
    return m


def main() -> None:
    sys.path.insert(0, str(GOLDEN.parent.parent))
    from cosmos_curate_amd.models.transnetv2 import make_transnetv2_weights

    ref_mod = load_reference_transnet()
    net = ref_mod._TransNetV2()
    missing, unexpected = net.load_state_dict(make_transnetv2_weights(), strict=True), None
    net.eval()

    rng = np.random.default_rng(0x7A45)
    x = rng.integers(0, 256, size=(1, 100, 27, 48, 3), dtype=np.uint8)
    with torch.no_grad():
        out = net(torch.from_numpy(x))
    np.savez_compressed(
        GOLDEN / "transnetv2_golden.npz",
        input=x,
        output=out[0, :, 0].numpy().astype(np.float32),
    )
    print("wrote transnetv2 golden:", out.mean().item())


if __name__ == "__main__":
    main()
