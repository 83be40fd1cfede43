"""In-tree build of the gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built parallel_convolution_amd/_C*.so travels to the GPU box with the
gpurun snapshot (it is git-ignored but NOT gpurun-ignored).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROCM_LIB = os.path.join(os.environ.get("ROCM_PATH", "/opt/rocm"), "lib")

# Debug build: PCONV_ASAN=1 adds host AddressSanitizer
# (run python under LD_PRELOAD=$(gcc -print-file-name=libasan.so)) and -g.
_asan = bool(os.environ.get("PCONV_ASAN"))
_dbg_cxx = ["-fsanitize=address", "-g", "-fno-omit-frame-pointer"] if _asan else []
_dbg_ld = ["-fsanitize=address"] if _asan else []

ext = CUDAExtension(
    name="parallel_convolution_amd._C",
    sources=[
        "csrc/bindings.cpp",
        "csrc/conv_kernels.hip",
        "csrc/io.cpp",
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"] + _dbg_cxx,
        "nvcc": ["-O3", "-std=c++17"],
    },
    libraries=["rccl"],
    library_dirs=[ROCM_LIB],
    extra_link_args=[f"-Wl,-rpath,{ROCM_LIB}"] + _dbg_ld,
)

setup(
    name="parallel_convolution_amd",
    version="0.1.0",
    packages=["parallel_convolution_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension},
)
