"""Build the ddstore_amd native extension in-tree.

Usage:  python setup.py build_ext --inplace

Compiles the CDNA4 (gfx950) HIP sources with hipcc via torch's extension
machinery. `.hip` sources are compiled directly (no hipify pass). The built
`ddstore_amd/_C*.so` lives in-tree so it travels with repo snapshots.
"""
import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))

setup(
    name="ddstore_amd",
    version="0.1.0",
    description="MI355X-native distributed in-HBM sample store (DDStore capabilities)",
    packages=["ddstore_amd"],
    py_modules=["pyddstore"],
    ext_modules=[
        CUDAExtension(
            name="ddstore_amd._C",
            sources=[
                "ddstore_amd/csrc/ddstore_core.hip",
                "ddstore_amd/csrc/ddstore_kernels.hip",
            ],
            include_dirs=[os.path.join(ROOT, "ddstore_amd", "csrc")],
            libraries=["roctx64"],
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
