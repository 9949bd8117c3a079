"""lambdipy_amd — an MI355X-native scientific-Python packager + runtime.

A from-scratch framework with the capabilities of the reference packager
(customink/lambdipy: CLI, build-recipe format, dependency resolver, packaged
artifact assembly — see SURVEY.md §1.1/§2.1; the reference mount was empty, so
parity targets are cited against SURVEY.md/BASELINE.json, not file:line), plus
an MI355X-native runtime half the reference does not have:

- per-package recipes emit ROCm wheels whose hot ops (GEMM, batched GEMV, FFT,
  CSR SpMV, pairwise-distance, RandomForest traversal) are hand-written CDNA4
  HIP kernels (gfx950, MFMA + LDS tiling) — ``lambdipy_amd.ops``;
- the packager strips/compresses to a Lambda-layer-sized zip artifact —
  ``lambdipy_amd.packager``;
- a cold-start runtime that times unzip → import → first inference on MI355X —
  ``lambdipy_amd.runtime``;
- whole-node batch fan-out over RCCL broadcast/all-gather on xGMI —
  ``lambdipy_amd.parallel``.

No CUDA-compat shims, no CUDA/HIP dual paths, no Triton, no multi-backend
dispatch: gfx950 only.
"""

__version__ = "0.1.0"
