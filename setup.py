"""Build the in-tree CDNA4 (gfx950) extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The .so lands in openembedding_amd/ops/_embops*.so (in-tree, so it travels
with the repo snapshot to GPU boxes; it is git-ignored)."""

import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))

ext = CUDAExtension(
    name="openembedding_amd.ops._embops",
    sources=[
        "openembedding_amd/ops/csrc/bindings.cpp",
        "openembedding_amd/ops/csrc/embops.hip",
        "openembedding_amd/ops/csrc/ctrhead.hip",
        "openembedding_amd/ops/csrc/mlp.hip",
        "openembedding_amd/ops/csrc/cin.hip",
    ],
    extra_compile_args={
        "cxx": ["-O3"],
        # hipcc flags; -ffp-contract=off keeps optimizer/init math op-for-op
        # identical to the torch float32 oracle (no fma contraction).
        "nvcc": ["-O3", "-ffp-contract=off"],
    },
)

setup(
    name="openembedding_amd",
    version="0.2.0",
    packages=find_packages(include=["openembedding_amd",
                                    "openembedding_amd.*"]),
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension},
)
