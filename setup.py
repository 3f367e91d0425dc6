"""Build the CDNA4 (gfx950) HIP extension in-tree:

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands next to the package (hivemind_amd/ops/) so it travels with
the repo snapshot to GPU boxes. hipcc cross-compiles without a GPU present.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402 (after env set)

this_dir = os.path.dirname(os.path.abspath(__file__))

ext = CUDAExtension(
    name="hivemind_amd._hip_ops",
    sources=[
        "hivemind_amd/ops/hip/bindings.hip",
        "hivemind_amd/ops/hip/mfma_gemm_impl.hip",
        "hivemind_amd/ops/hip/flash_attention.hip",
        "hivemind_amd/ops/hip/multi_tensor.hip",
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17"],
    },
)

setup(
    name="hivemind_amd",
    version="0.1.0",
    packages=["hivemind_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=True)},
)
