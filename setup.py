"""In-tree build of the genrec_amd HIP extension for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built genrec_amd/_C*.so lives in the package so it travels with the
repo snapshot to GPU boxes (no JIT cache dependency).
"""

import os

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

ROOT = os.path.dirname(os.path.abspath(__file__))

sources = [
    "csrc/bindings.cpp",
    "csrc/kernels/norms.hip",
    "csrc/kernels/attention.hip",
    "csrc/kernels/attention_mfma.hip",
    "csrc/kernels/hstu_attn.hip",
    "csrc/kernels/ce.hip",
    "csrc/kernels/quantize.hip",
    "csrc/kernels/metrics.hip",
    "csrc/kernels/embedding.hip",
    "csrc/kernels/fused_elementwise.hip",
    "csrc/kernels/adamw.hip",
    "csrc/kernels/attention_flash.hip",
    "csrc/kernels/skinny_gemm.hip",
]

cxx_flags = ["-O3", "-std=c++17"]
hip_flags = ["-O3", "-std=c++17"]
if os.environ.get("GENREC_DEBUG_BUILD", "0") == "1":
    # debug / sanitizer-friendly build: symbols, no
    # aggressive opt, frame pointers for rocgdb
    cxx_flags = ["-O1", "-g", "-std=c++17", "-fno-omit-frame-pointer"]
    hip_flags = ["-O1", "-g", "-std=c++17", "-fno-omit-frame-pointer"]

from setuptools import find_packages

setup(
    name="genrec_amd",
    version="0.1.0",
    description=("MI355X-native generative recommendation framework "
                 "(CDNA4 HIP kernels + PyTorch-ROCm + RCCL)"),
    packages=find_packages(include=["genrec_amd", "genrec_amd.*"]),
    python_requires=">=3.10",
    ext_modules=[
        CUDAExtension(
            name="genrec_amd._C",
            sources=sources,
            extra_compile_args={"cxx": cxx_flags, "nvcc": hip_flags},
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
