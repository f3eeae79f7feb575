"""Build the in-tree CDNA4 (gfx950) kernel extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The .so lands next to its package (calfkit_amd/ops/_C*.so) so the repo
snapshot carries it to GPU boxes; the JIT cache never gets involved.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "calfkit_amd", "ops", "csrc")

sources = [
    os.path.join(CSRC, "bindings.cpp"),
    os.path.join(CSRC, "elementwise.hip"),
    os.path.join(CSRC, "attn_decode.hip"),
    os.path.join(CSRC, "attn_prefill.hip"),
    os.path.join(CSRC, "sampling.hip"),
    os.path.join(CSRC, "skinny_gemm.hip"),
    os.path.join(CSRC, "tile_gemm.hip"),
    os.path.join(CSRC, "moe_gemm.hip"),
    os.path.join(CSRC, "gemm_search.hip"),
]

setup(
    name="calfkit-amd-kernels",
    ext_modules=[
        CUDAExtension(
            name="calfkit_amd.ops._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
            libraries=["hipblaslt"],
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
