"""In-tree build of the amdtrain gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The resulting amdtrain/_C.*.so stays in the source tree so it travels with
repo snapshots to the GPU box (no JIT cache dependence).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

CSRC = os.path.join("amdtrain", "ops", "csrc")

ext = CUDAExtension(
    name="amdtrain._C",
    sources=[
        os.path.join(CSRC, "ext.cpp"),
        os.path.join(CSRC, "sgd.hip"),
        os.path.join(CSRC, "cross_entropy.hip"),
        os.path.join(CSRC, "elementwise.hip"),
        os.path.join(CSRC, "batchnorm.hip"),
        os.path.join(CSRC, "gemm.hip"),
        os.path.join(CSRC, "conv3x3.hip"),
        os.path.join(CSRC, "conv_stem.hip"),
        os.path.join(CSRC, "gemm8p.hip"),
        os.path.join(CSRC, "wgrad.hip"),
        os.path.join(CSRC, "pool.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17"],
    },
)

setup(
    name="amdtrain",
    version="0.1.0",
    packages=["amdtrain", "amdtrain.utils", "amdtrain.models", "amdtrain.ops",
              "amdtrain.parallel", "amdtrain.comm", "amdtrain.data",
              "amdtrain.engine", "amdtrain.cli"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension},
)
