"""Build the in-tree HIP extension for gfx950.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built ``rayfed_amd/_hip*.so`` is git-ignored but travels with repo
snapshots to GPU boxes (it must live in-tree, not in site-packages).
"""
import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils import cpp_extension  # noqa: E402

setup(
    name="rayfed_amd",
    version="0.2.0",
    packages=find_packages(include=["rayfed_amd", "rayfed_amd.*"]),
    ext_modules=[
        cpp_extension.CUDAExtension(
            name="rayfed_amd._hip",
            sources=["csrc/rayfed_hip.hip"],
            include_dirs=[os.path.abspath("csrc")],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        ),
        cpp_extension.CppExtension(
            name="rayfed_amd._xfer",
            sources=["csrc/xfer_core.cpp"],
            extra_compile_args=["-O3", "-std=c++17", "-pthread"],
            # amdhip64: striped chunk bodies assemble into hipHostMalloc'd
            # (pinned) buffers so the consumer H2Ds them zero-copy; falls
            # back to plain malloc when no GPU is present.
            libraries=["ssl", "crypto", "amdhip64"],
            library_dirs=["/opt/rocm/lib"],
            include_dirs=["/opt/rocm/include"],
        ),
    ],
    cmdclass={"build_ext": cpp_extension.BuildExtension},
)
