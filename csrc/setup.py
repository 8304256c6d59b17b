"""AOT build of the gansformer_amd._C HIP extension for gfx950.

Usage (from the repo root):
    PYTORCH_ROCM_ARCH=gfx950 python csrc/setup.py build_ext --inplace

The built .so lands in gansformer_amd/ (in-tree, so gpurun snapshots
carry it to the GPU box). No import-time JIT.
"""

import os
import sys

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

CSRC = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(CSRC)

sources = [os.path.join(CSRC, f) for f in (
    "ext.hip", "fba.hip", "upfirdn2d.hip", "conv2d.hip", "conv2d_slab.hip",
    "conv2d_wgrad_slab.hip", "conv2d_up2.hip", "conv2d_s2.hip",
    "upfirdn2d_sep.hip", "modnorm.hip",
    "mbstd.hip", "attn.hip", "attn_bwd.hip", "gemm_skinny.hip",
    "pack.hip")]

setup(
    name="gansformer_amd_ext",
    ext_modules=[
        CUDAExtension(
            name="gansformer_amd._C",
            sources=sources,
            include_dirs=[CSRC],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
    script_args=sys.argv[1:] or ["build_ext", "--inplace"],
)
