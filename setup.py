"""Build for the in-tree gfx950 HIP extension.

Usage: `python setup.py build_ext --inplace` (hipcc cross-compiles without a
GPU).  The resulting `distributed_embeddings_amd/_hip_ops*.so` is loaded by
`distributed_embeddings_amd.ops._backend` and travels with the source tree.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ext = CUDAExtension(
    name="distributed_embeddings_amd._hip_ops",
    sources=[
        "distributed_embeddings_amd/csrc/bindings.cpp",
        "distributed_embeddings_amd/csrc/embedding_ops.hip",
        "distributed_embeddings_amd/csrc/dot_interact.hip",
        "distributed_embeddings_amd/csrc/radix_sort.hip",
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17"],
    },
)

setup(
    name="distributed_embeddings_amd",
    version="0.1.0",
    packages=[
        "distributed_embeddings_amd",
        "distributed_embeddings_amd.ops",
        "distributed_embeddings_amd.layers",
        "distributed_embeddings_amd.parallel",
        "distributed_embeddings_amd.models",
        "distributed_embeddings_amd.utils",
    ],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension},
)
