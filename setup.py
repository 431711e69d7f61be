"""Build the in-tree HIP extension (quintnet_amd._C) for gfx950.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

hipcc cross-compiles without a GPU; the resulting .so lives in-tree so
it travels with the repo snapshot to GPU boxes.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

HERE = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(HERE, "quintnet_amd", "csrc")

sources = [
    os.path.join(CSRC, "bindings.cpp"),
    os.path.join(CSRC, "layernorm.hip"),
    os.path.join(CSRC, "softmax.hip"),
    os.path.join(CSRC, "cross_entropy.hip"),
    os.path.join(CSRC, "adamw.hip"),
    os.path.join(CSRC, "elementwise.hip"),
    os.path.join(CSRC, "attn.hip"),
    os.path.join(CSRC, "embedding.hip"),
    os.path.join(CSRC, "gemm.hip"),
    os.path.join(CSRC, "wgrad.hip"),
    os.path.join(CSRC, "blaslt.cpp"),
]

setup(
    name="quintnet_amd",
    version="0.1.0",
    packages=["quintnet_amd"],
    ext_modules=[
        CUDAExtension(
            name="quintnet_amd._C",
            sources=sources,
            libraries=["hipblaslt"],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950",
                         "-mllvm", "-amdgpu-mfma-vgpr-form"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
