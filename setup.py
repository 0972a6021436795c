"""Build the in-tree HIP extension for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built bigslice_amd/_C*.so travels with the repo snapshot to GPU boxes;
there is no JIT cache dependency.
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup
from torch.utils import cpp_extension

ROOT = os.path.dirname(os.path.abspath(__file__))

ext = cpp_extension.CUDAExtension(
    name="bigslice_amd._C",
    sources=["bigslice_amd/csrc/ext.hip"],
    include_dirs=[os.path.join(ROOT, "bigslice_amd/csrc")],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17"],
    },
)

setup(
    name="bigslice_amd",
    version="0.1.0",
    packages=["bigslice_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": cpp_extension.BuildExtension},
)
