"""Build the relora_amd package and its in-tree gfx950 HIP extension.

Usage:  python setup.py build_ext --inplace
The built .so lands at relora_amd/ops/_relora_hip*.so and ships with the
repo snapshot (it is git-ignored but not gpurun-ignored).
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import find_packages, setup  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402
import pybind11  # noqa: E402
from setuptools import Extension  # noqa: E402

CSRC = os.path.join("relora_amd", "ops", "csrc")
DATA_CSRC = os.path.join("relora_amd", "data", "csrc")

# Host-side C++ index builders (no HIP): plain pybind11 extension.
index_helpers_ext = Extension(
    name="relora_amd.data._index_helpers",
    sources=[os.path.join(DATA_CSRC, "index_helpers.cpp")],
    include_dirs=[pybind11.get_include()],
    extra_compile_args=["-O3", "-std=c++17", "-fvisibility=hidden"],
    language="c++",
)

ext = CUDAExtension(
    name="relora_amd.ops._relora_hip",
    sources=[
        os.path.join(CSRC, "bindings.cpp"),
        os.path.join(CSRC, "norms.hip"),
        os.path.join(CSRC, "rope.hip"),
        os.path.join(CSRC, "swiglu.hip"),
        os.path.join(CSRC, "ce.hip"),
        os.path.join(CSRC, "adamw.hip"),
        os.path.join(CSRC, "attention.hip"),
        os.path.join(CSRC, "lora_gemm.hip"),
        os.path.join(CSRC, "fused_gemm.hip"),
        os.path.join(CSRC, "quantize.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        # RELORA_AMD_ROT_V2=1 at build time compiles the Q_V2 table-driven
        # LDS rotation (bank-model-verified conflict-free variant) for A/B;
        # default is the shipped t_rot/rot8 rotation.
        "nvcc": ["-O3", "-std=c++17"]
        + (["-DRELORA_AMD_ROT_V2=1"]
           if os.environ.get("RELORA_AMD_ROT_V2") == "1" else []),
    },
)

setup(
    name="relora_amd",
    version="0.1.0",
    packages=find_packages(include=["relora_amd", "relora_amd.*"]),
    ext_modules=[ext, index_helpers_ext],
    cmdclass={"build_ext": BuildExtension},
)
