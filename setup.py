"""Build the in-tree gfx950 HIP extension:

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces bdbnn_amd/_native*.so (git-ignored; travels with the repo
snapshot to the GPU box).  hipcc cross-compiles fine on a machine with
no GPU.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "bdbnn_amd", "csrc")

sources = [os.path.join(CSRC, f) for f in (
    "bind.cpp",
    "pack.hip",
    "prelu.hip",
    "bn_act.hip",
    "pool.hip",
    "loss.hip",
    "conv_dgrad.hip",
    "conv_bwd.hip",
    "conv_wgrad.hip",
    "stem_conv.hip",
    "xnor_conv.hip",
    "kurtosis.hip",
    "kd.hip",
    "optim.hip",
)]

setup(
    name="bdbnn_amd_native",
    ext_modules=[
        CUDAExtension(
            name="bdbnn_amd._native",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                # rocWMMA (conv_dgrad.hip) needs HIP half conversions,
                # which torch's extension defaults disable
                "nvcc": ["-O3", "-std=c++17",
                         "-U__HIP_NO_HALF_CONVERSIONS__",
                         "-U__HIP_NO_HALF_OPERATORS__"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
