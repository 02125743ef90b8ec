"""In-tree build of the gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python3 setup.py build_ext --inplace

Builds adversarial_spec_amd/ops/_advspec_hip.*.so next to its package so
the .so travels with the source tree (gpurun snapshots, no JIT cache).
hipcc cross-compiles without a GPU present.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

HIP_DIR = os.path.join("adversarial_spec_amd", "ops", "hip")

ext = CUDAExtension(
    name="adversarial_spec_amd.ops._advspec_hip",
    sources=[
        os.path.join(HIP_DIR, "bindings.hip"),
        os.path.join(HIP_DIR, "elementwise.hip"),
        os.path.join(HIP_DIR, "attention.hip"),
        os.path.join(HIP_DIR, "attn_prefill_mfma.hip"),
        os.path.join(HIP_DIR, "gemv.hip"),
        os.path.join(HIP_DIR, "gemm.hip"),
        os.path.join(HIP_DIR, "gemm256.hip"),
        os.path.join(HIP_DIR, "gemm_fp8.hip"),
        os.path.join(HIP_DIR, "gemm256_fp8.hip"),
        os.path.join(HIP_DIR, "sampling.hip"),
        os.path.join(HIP_DIR, "mfma_rate.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="adversarial-spec-amd-ext",
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension},
)
