"""In-tree build for the HIP extension (gfx950).

Usage: python setup.py build_ext --inplace
The built .so stays in-tree so the gpurun snapshot ships it.
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

ext_modules = []
cmdclass = {}
try:
    from torch.utils import cpp_extension

    ext_modules = [
        cpp_extension.CUDAExtension(
            name="distributedkernelshap_amd.ops._kshap_hip",
            sources=[
                "distributedkernelshap_amd/ops/hip/bindings.cpp",
                "distributedkernelshap_amd/ops/hip/kshap_kernels.hip",
            ],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "--offload-arch=gfx950", "-std=c++17"],
            },
        )
    ]
    cmdclass = {"build_ext": cpp_extension.BuildExtension}
except Exception:
    pass

setup(
    name="distributedkernelshap_amd",
    version="0.1.0",
    packages=[
        "distributedkernelshap_amd",
        "distributedkernelshap_amd.core",
        "distributedkernelshap_amd.explainers",
        "distributedkernelshap_amd.models",
        "distributedkernelshap_amd.ops",
        "distributedkernelshap_amd.parallel",
        "distributedkernelshap_amd.serve",
        "distributedkernelshap_amd.utils",
    ],
    ext_modules=ext_modules,
    cmdclass=cmdclass,
)
