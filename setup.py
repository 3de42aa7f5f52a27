"""Build the in-tree HIP extension for gfx950.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built gcbf_amd/_C*.so stays in-tree so it ships with repo snapshots.
"""
import os

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

HIP_DIR = os.path.join("gcbf_amd", "ops", "hip")

setup(
    name="gcbf_amd_ext",
    ext_modules=[
        CUDAExtension(
            name="gcbf_amd._C",
            sources=sorted(
                os.path.join(HIP_DIR, f) for f in os.listdir(HIP_DIR)
                if f.endswith((".cpp", ".hip"))),
            extra_compile_args={
                "cxx": ["-O3"],
                "nvcc": ["-O3", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
