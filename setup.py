"""In-tree HIP-native build of the gfx950 extension.

    python setup.py build_ext --inplace

Every translation unit is compiled by hipcc for gfx950 directly — there
is no hipify pass and no CUDA-compat layer anywhere in the pipeline
(BASELINE.json:5 "no CUDA-compat shims"): csrc/ is written against the
HIP/ROCm APIs natively, and bindings.cpp uses only platform-neutral
torch C++ (tensors, pybind) plus the HIP runtime.

The built parallel_convolution_amd/_C*.so travels to the GPU box with
the gpurun snapshot (git-ignored but NOT gpurun-ignored).

PCONV_ASAN=1 adds host AddressSanitizer + -g; run
python under LD_PRELOAD=$(gcc -print-file-name=libasan.so).
"""

import os
import subprocess
import sys
import sysconfig

from setuptools import Extension, find_packages, setup
from setuptools.command.build_ext import build_ext

ROCM = os.environ.get("ROCM_PATH", "/opt/rocm")
HIPCC = os.path.join(ROCM, "bin", "hipcc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950").split(";")[0]

import torch  # noqa: E402  (include paths + ABI flag come from the torch build)
from torch.utils.cpp_extension import include_paths, library_paths  # noqa: E402

SOURCES = ["csrc/bindings.cpp", "csrc/conv_kernels.hip", "csrc/io.cpp"]

_asan = bool(os.environ.get("PCONV_ASAN"))


def _common_flags():
    flags = [
        "-O3",
        "-std=c++17",
        "-fPIC",
        f"--offload-arch={ARCH}",
        "-fno-gpu-rdc",
        "-DUSE_ROCM=1",
        f"-D_GLIBCXX_USE_CXX11_ABI={int(torch.compiled_with_cxx11_abi())}",
        "-DTORCH_EXTENSION_NAME=_C",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
    ]
    if _asan:
        flags += ["-fsanitize=address", "-g", "-fno-omit-frame-pointer"]
    for d in include_paths() + [sysconfig.get_paths()["include"]]:
        flags.append(f"-I{d}")
    flags.append(f"-I{os.path.join(ROCM, 'include')}")
    return flags


class HipccBuildExt(build_ext):
    """Compile + link with hipcc directly (no hipify, no nvcc emulation)."""

    def build_extension(self, ext):
        out_path = self.get_ext_fullpath(ext.name)
        os.makedirs(os.path.dirname(out_path), exist_ok=True)
        objdir = os.path.join(self.build_temp, "obj")
        os.makedirs(objdir, exist_ok=True)

        flags = _common_flags()
        dep_mtime = max(
            os.path.getmtime(p)
            for p in ["setup.py"]
            + [os.path.join("csrc", f) for f in os.listdir("csrc")]
        )
        objs = []
        for src in ext.sources:
            obj = os.path.join(
                objdir, os.path.basename(src).replace("/", "_") + ".o"
            )
            objs.append(obj)
            if os.path.exists(obj) and os.path.getmtime(obj) > dep_mtime:
                continue
            cmd = [HIPCC, "-c", src, "-o", obj] + flags
            print("[hipcc]", " ".join(cmd), flush=True)
            subprocess.run(cmd, check=True)

        torch_lib = library_paths()[0]
        rocm_lib = os.path.join(ROCM, "lib")
        link = [
            HIPCC,
            "-shared",
            "-fPIC",
            *objs,
            "-o",
            out_path,
            f"-L{torch_lib}",
            "-lc10",
            "-ltorch",
            "-ltorch_cpu",
            "-ltorch_python",
            "-lc10_hip",
            "-ltorch_hip",
            f"-L{rocm_lib}",
            "-lrccl",
            "-lamdhip64",
            f"-Wl,-rpath,{torch_lib}",
            f"-Wl,-rpath,{rocm_lib}",
        ]
        if _asan:
            link.append("-fsanitize=address")
        print("[hipcc-link]", " ".join(link), flush=True)
        subprocess.run(link, check=True)


setup(
    name="parallel_convolution_amd",
    version="0.2.0",
    packages=find_packages(include=["parallel_convolution_amd*"]),
    ext_modules=[Extension("parallel_convolution_amd._C", sources=SOURCES)],
    cmdclass={"build_ext": HipccBuildExt},
)
