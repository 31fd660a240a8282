"""Build the in-tree HIP extension for gfx950 (MI355X).

  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

hipcc cross-compiles without a GPU; the built .so lives in
code2vec_amd/ops/ and travels with the repo snapshot to GPU boxes.
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CppExtension, CUDAExtension

SRC = [
    "code2vec_amd/ops/csrc/bindings.cpp",
    "code2vec_amd/ops/csrc/gather_concat.hip",
    "code2vec_amd/ops/csrc/combiner.hip",
    "code2vec_amd/ops/csrc/attention.hip",
    "code2vec_amd/ops/csrc/logsoftmax_nll.hip",
    "code2vec_amd/ops/csrc/adam.hip",
    "code2vec_amd/ops/csrc/wgrad.hip",
    "code2vec_amd/ops/csrc/colsum.hip",
    "code2vec_amd/ops/csrc/head_fwd.hip",
    "code2vec_amd/ops/csrc/head_dgrad.hip",
    "code2vec_amd/ops/csrc/head_bwd.hip",
    "code2vec_amd/ops/csrc/angular.hip",
    "code2vec_amd/ops/csrc/dgrad2.hip",
    "code2vec_amd/ops/csrc/dgrad.hip",
    "code2vec_amd/ops/csrc/head_wgrad.hip",
]

setup(
    name="code2vec_amd_ext",
    ext_modules=[
        CppExtension(
            name="code2vec_amd.data._c2v_host",
            sources=["code2vec_amd/data/csrc/epoch_builder.cpp"],
            extra_compile_args=["-O3", "-std=c++17", "-fopenmp"],
            extra_link_args=["-fopenmp"],
        ),
        CppExtension(
            name="code2vec_amd.data._c2v_extract",
            sources=["code2vec_amd/data/csrc/extractor.cpp"],
            extra_compile_args=["-O3", "-std=c++17"],
        ),
        CUDAExtension(
            name="code2vec_amd.ops._c2v_hip",
            sources=SRC,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
