"""cosmos_curate_amd: MI355X-native rebuild of cosmos-curate's video hot path.

Scope: the per-clip decode -> frame-sample -> preprocess ->
embed path behind the reference's ``cosmos_curate.core.interfaces``
Stage/Model plugin surface, built from scratch for gfx950 (CDNA4):

- host side: a byte-compatible mirror of the plugin surface
  (``cosmos_curate_amd.core.interfaces``), the video data model, and the
  split/embedding stages (``cosmos_curate_amd.pipelines.video``);
- device side: hand-written HIP kernels (NV12->RGB, resize, CLIP
  preprocess, bf16 MFMA GEMMs) behind a C ABI (``include/cc_hotpath.h``,
  implemented in ``cosmos_curate_amd/csrc``), bound via ctypes in
  ``cosmos_curate_amd.hotpath``.

The product path NEVER falls back to CPU compute on a GPU box: if the HIP
extension is missing, GPU stages raise (see hotpath.require()).
"""

__version__ = "0.1.0"
