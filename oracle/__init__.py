"""CPU oracle for the cosmos-curate video split/annotate hot path.

TEST INFRASTRUCTURE ONLY.  This package is a plain-numpy / torch-cpu
restatement of the reference pipeline's hot-path arithmetic
(nvidia-cosmos/cosmos-curate, mounted read-only at /root/reference during
development).  It exists so the MI355X HIP product path in
``cosmos_curate_amd`` can be parity-tested against a pinned CPU definition.

Only ``tests/``, ``__graft_entry__.smoke()`` and ``bench.py``'s
``cpu_baseline`` leg may import or call anything in here.  The product path
never routes through this package: on a GPU box the HIP extension either
loads or the product path raises.

Parity pinning status:
- sampling (find_closest_indices / sample_closest): pinned by the reference's
  own known-answer tables (tests/cosmos_curate/pipelines/video/utils/
  test_decoder_utils.py:40-201), replicated as data in tests/golden/.
- span math + clip UUIDs: pinned by the reference's fixed-stride tests
  (test_fixed_stride_extraction.py:100-320) replicated as data.
- MP4 demux -> PTS: pinned against the reference's Sintel fixtures
  (test_clip_10s.mp4 / test_video_30s.mp4), parsed in the dev container; the
  derived PTS arrays are committed under tests/golden/ with the generating
  script.
- NV12->RGB / resize pixel arithmetic: the reference delegates to
  CVCUDA/cv2 whose pixel output its own tests never pin bit-exactly
  (SURVEY.md §8c); our oracle restates the published BT.601 /
  bilinear / Catmull-Rom(a=-0.75) formulas and is the definition the HIP
  kernels are tested against.  End-to-end pixel parity with the reference is
  pinned via embedding cosine.
- ViT (CLIP-ViT-B/32): torch-cpu fp32 forward with fixed-seed weights
  (HF weights unavailable offline) -- numerics oracle for the bf16 MFMA path.
"""
