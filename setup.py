"""In-tree build of the MI355X-native extensions (gfx950 only).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces dear_pytorch_amd/_comm_core*.so (RCCL communicator) and
dear_pytorch_amd/_kernels*.so (fused CDNA4 kernels).  Built .so files live
in-tree so the gpurun snapshot carries them to the GPU box.
"""
import os

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension, ROCM_HOME

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

rocm_lib = os.path.join(ROCM_HOME or "/opt/rocm", "lib")
rocm_inc = os.path.join(ROCM_HOME or "/opt/rocm", "include")

common = dict(
    include_dirs=[rocm_inc],
    library_dirs=[rocm_lib],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17"],
    },
)

ext_modules = [
    CUDAExtension(
        name="dear_pytorch_amd._comm_core",
        sources=["dear_pytorch_amd/csrc/comm_core.cpp"],
        libraries=["rccl", "amdhip64"],
        **common,
    ),
    CUDAExtension(
        name="dear_pytorch_amd._kernels",
        sources=["dear_pytorch_amd/csrc/kernels.hip",
                 "dear_pytorch_amd/csrc/bn_kernels.hip"],
        libraries=["amdhip64"],
        **common,
    ),
]

setup(
    name="dear_pytorch_amd",
    version="0.1.0",
    packages=["dear_pytorch_amd", "dear_pytorch_amd.comm",
              "dear_pytorch_amd.parallel", "dear_pytorch_amd.ops",
              "dear_pytorch_amd.models", "dear_pytorch_amd.utils"],
    ext_modules=ext_modules,
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
