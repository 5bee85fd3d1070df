"""In-tree build of the HIP/CDNA4 extension for gfx950.

    python setup.py build_ext --inplace

produces flink_parameter_server_amd/_C*.so (git-ignored; the .so still
travels to the GPU box with the repo snapshot).
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="flink_parameter_server_amd",
    version="0.1.0",
    packages=["flink_parameter_server_amd"],
    ext_modules=[
        CUDAExtension(
            name="flink_parameter_server_amd._C",
            sources=["csrc/ps_kernels.hip"],
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950", "-ffp-contract=off"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
