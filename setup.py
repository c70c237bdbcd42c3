"""In-tree build of the HIP/CDNA4 extension (gfx950 only).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands next to the package (deepreduce_amd/_hip_ops*.so) so it
travels with the repo snapshot to GPU boxes.
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="deepreduce_amd",
    version="0.1.0",
    packages=["deepreduce_amd"],
    ext_modules=[
        CUDAExtension(
            name="deepreduce_amd._hip_ops",
            sources=["deepreduce_amd/ops/src/hip_ops.hip"],
            extra_compile_args={"cxx": ["-O3"], "nvcc": ["-O3"]},
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
