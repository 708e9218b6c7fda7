"""spectre_amd — MI355X-native BN254 MSM/NTT proving backend for the
Spectre/halo2 hot path.

The product surface is the C ABI in include/spectre_gpu.h, implemented by
libspectre_gpu.so (hand-written HIP kernels for gfx950). This package is the
Python-side mirror of that boundary, used by tests and the bench harness;
the real consumer is a patched halo2_proofs calling the C ABI directly
(INTEGRATION.md). There is NO CPU fallback: every compute call requires an
AMD GPU and fails loudly without one.
"""
from .ffi import SpectreGpu, lib_path, load_library  # noqa: F401

__version__ = "0.1.0"
