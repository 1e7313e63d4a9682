"""Build the in-tree HIP extension for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built _bee2bee_hip.so lands inside bee2bee_amd/ops/ so it travels with
the repo snapshot to GPU boxes (no JIT cache dependency).
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

CSRC = os.path.join("bee2bee_amd", "ops", "csrc")

ext = CUDAExtension(
    name="bee2bee_amd.ops._bee2bee_hip",
    sources=[
        os.path.join(CSRC, "ext.cpp"),
        os.path.join(CSRC, "elementwise.hip"),
        os.path.join(CSRC, "attn_decode.hip"),
        os.path.join(CSRC, "attn_prefill.hip"),
        os.path.join(CSRC, "probe.hip"),
        os.path.join(CSRC, "grouped_gemm.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"]
        + (["-DBEE2BEE_DEBUG"] if os.environ.get("BEE2BEE_DEBUG_KERNELS") == "1" else []),
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"]
        + (["-DBEE2BEE_DEBUG", "-g"] if os.environ.get("BEE2BEE_DEBUG_KERNELS") == "1" else []),
    },
)

setup(
    name="bee2bee-amd-ext",
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
