"""Build the in-tree native extension: python setup.py build_ext --inplace.

Compiles the CDNA4 HIP kernel library for gfx950 via hipcc (works without
a GPU — cross-compile). The resulting audiomuse_amd/_C.*.so stays in-tree
so it travels with repo snapshots.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

CSRC = os.path.join("audiomuse_amd", "ops", "csrc")

sources = [
    os.path.join(CSRC, "bindings.cpp"),
    os.path.join(CSRC, "gemm_gelu.cpp"),
    os.path.join(CSRC, "mel.hip"),
]
for extra in ("distance.hip", "kmeans.hip", "attention.hip", "norms.hip"):
    p = os.path.join(CSRC, extra)
    if os.path.exists(p):
        sources.append(p)

setup(
    name="audiomuse-amd",
    version="0.1.0",
    packages=["audiomuse_amd"],
    ext_modules=[
        CUDAExtension(
            "audiomuse_amd._C",
            sources,
            libraries=["hipblaslt"],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
